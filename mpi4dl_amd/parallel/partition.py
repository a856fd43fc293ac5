"""Model partitioner + boundary shape inference.

Reference parity: torchgems ``model_generator`` (src/torchgems/
mp_pipeline.py:41-168). Differences by design:

* Shape inference runs the model on the **meta device** at the real
  image size — free, exact, and layer-shape-general — instead of the
  reference's trick of a dummy CUDA forward at a small "seq" size whose
  shapes are then rescaled (mp_pipeline.py:126-168, train_spatial.py:61-238),
  which silently breaks on resolution-inhomogeneous layers.
* ``ready_model`` moves ONLY the local partition's weights to the
  target device (and drops the host-side copy of remote partitions);
  the reference materialises the full model on every rank's GPU.

Spatial/halo modules in this package are meta-aware: on meta tensors
they skip communication and produce correctly-shaped outputs, so the
same inference path covers spatial models.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple, Union

import torch
import torch.nn as nn

ShapeSpec = Union[Tuple[int, ...], List[Tuple[int, ...]]]


def _shapes_of(y) -> ShapeSpec:
    if isinstance(y, (tuple, list)):
        return [tuple(t.shape) for t in y]
    return tuple(y.shape)


class model_generator:
    """Split a flat ``nn.Sequential`` into ``split_size`` contiguous stages.

    Parameters
    ----------
    model : nn.Sequential built at cell granularity (models/ builders).
    split_size : number of pipeline stages.
    input_size : full input shape (N, C, H, W) used for shape inference.
        N may be the micro-batch ("parts") size; shapes are per micro-batch.
    balance : optional per-stage layer counts (sums to len(model));
        reference mp_pipeline.py:41-83.
    """

    def __init__(
        self,
        model: nn.Sequential,
        split_size: int,
        input_size: Tuple[int, ...],
        balance: Optional[Sequence[int]] = None,
    ):
        assert isinstance(model, nn.Sequential), "partitioner expects nn.Sequential"
        self.model = model
        self.split_size = split_size
        self.input_size = tuple(input_size)
        n_layers = len(model)
        if balance is not None:
            balance = [int(b) for b in balance]
            assert len(balance) == split_size, (
                f"balance {balance} has {len(balance)} entries, need {split_size}"
            )
            assert sum(balance) == n_layers, (
                f"balance {balance} sums to {sum(balance)}, model has {n_layers} layers"
            )
            self.balance = balance
        else:
            base, rem = divmod(n_layers, split_size)
            self.balance = [base + (1 if i < rem else 0) for i in range(split_size)]
        # start/end layer index per stage
        self.bounds: List[Tuple[int, int]] = []
        s = 0
        for b in self.balance:
            self.bounds.append((s, s + b))
            s += b
        self.shape_list: List[ShapeSpec] = []  # output shapes per stage
        self.models: Optional[nn.Module] = None  # local stage after ready_model
        self.ready_rank: Optional[int] = None

    # -- reference-compat accessors -----------------------------------------

    def get_start_end_layer_index(self, split_rank: int) -> Tuple[int, int]:
        return self.bounds[split_rank]

    def get_model(self, split_rank: int) -> nn.Sequential:
        s, e = self.bounds[split_rank]
        return self.model[s:e]

    # -- shape inference -----------------------------------------------------

    def get_output_shapes(self, input_size: Optional[Tuple[int, ...]] = None):
        """Per-stage output shapes via one meta-device forward."""
        input_size = tuple(input_size or self.input_size)
        self.shape_list = infer_boundary_shapes(self.model, self.bounds, input_size)
        return self.shape_list

    # -- materialisation -----------------------------------------------------

    def ready_model(
        self,
        split_rank: int,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
    ) -> nn.Module:
        """Move the local stage to ``device``; infer shapes if not done."""
        if device is None:
            device = (
                torch.device("cuda", torch.cuda.current_device())
                if torch.cuda.is_available()
                else torch.device("cpu")
            )
        if not self.shape_list:
            self.get_output_shapes()
        local = self.get_model(split_rank)
        local = local.to(device=device, dtype=dtype)
        self.models = local
        self.ready_rank = split_rank
        # free the host copy of remote stages (shape inference is done;
        # at F416-scale the full model is ~0.5 GB/process of host RAM)
        self.model = None
        return local


@torch.no_grad()
def infer_boundary_shapes(
    model: nn.Sequential,
    bounds: Sequence[Tuple[int, int]],
    input_size: Tuple[int, ...],
) -> List[ShapeSpec]:
    """Output shape(s) of each stage for a given input shape.

    Runs the whole model once on the meta device — no FLOPs, no memory.
    Handles tuple activations (AmoebaNet passes (x, skip) between cells,
    reference amoebanet.py:449-533).
    """
    meta_model = _to_meta(model)
    x = torch.zeros(input_size, device="meta")
    shapes: List[ShapeSpec] = []
    for s, e in bounds:
        for i in range(s, e):
            x = meta_model[i](x)
        shapes.append(_shapes_of(x))
    return shapes


def _to_meta(model: nn.Module) -> nn.Module:
    """A structural copy of ``model`` with meta-device parameters/buffers.

    ProcessGroup handles (TileBatchNorm2d.group), HaloExchangers and
    other comm objects are passed through by identity — they are neither
    copyable nor needed for a meta forward (spatial modules skip comm on
    meta tensors).
    """
    import copy

    memo = {}
    try:
        from torch.distributed import ProcessGroup

        for m in model.modules():
            for v in m.__dict__.values():
                if isinstance(v, ProcessGroup):
                    memo[id(v)] = v
    except ImportError:
        pass
    from ..ops.halo import HaloExchanger

    for m in model.modules():
        for v in m.__dict__.values():
            if isinstance(v, HaloExchanger):
                memo[id(v)] = v
    meta = copy.deepcopy(model, memo)
    meta = meta.to(device="meta")
    return meta
