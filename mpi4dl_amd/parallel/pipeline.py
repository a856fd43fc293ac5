"""LP (layer parallelism) + PP (pipeline) training engine.

Reference parity: torchgems ``train_model`` (src/torchgems/
mp_pipeline.py:171-538): per-rank forward/backward over ``parts``
micro-batches with point-to-point activation/grad transfer between
stages, GPipe-style all-forwards-then-all-backwards schedule, tuple
(multi-tensor) activations, GEMS_INVERSE mirrored peer maps
(mp_pipeline.py:238-248).

MI355X-native design:
* P2P via grouped RCCL send/recv over xGMI (p2p.py), stream-ordered —
  no host-side synchronize fences;
* persistent preallocated recv buffers per micro-batch
  (mp_pipeline.py:250-292 keeps the same structure);
* optional bf16 autocast compute with a fixed boundary ``act_dtype`` so
  activation messages are half the bytes of the reference's fp32;
* loss scaled by 1/parts at backward so gradients are the mean over the
  full batch (the reference instead rescales grads in the allreduce,
  comm.py:440-458).
"""

from __future__ import annotations

import logging
import os
from typing import List, Optional

import torch
import torch.nn as nn

from .. import p2p
from ..comm import Communicator, current_device
from ..utils import GLOBAL_TIMER

log = logging.getLogger(__name__)


def _as_list(spec):
    """Shape spec -> list of shape tuples (single tensor = 1-list)."""
    if isinstance(spec, list):
        return [tuple(s) for s in spec]
    return [tuple(spec)]


class train_model:
    """One pipeline-stage engine living on one rank/GPU.

    Parameters mirror the reference where meaningful; ``comm`` supplies
    all topology. ``local_rank`` is the engine's in-clique position
    (NOT the global rank); for a GEMS-inverse engine it is the mirrored
    position and peers are mapped through ``comm.engine_peer``.
    """

    def __init__(
        self,
        model_gen,
        local_rank: int,
        batch_size: int,
        parts: int,
        comm: Communicator,
        criterion: Optional[nn.Module] = None,
        optimizer: Optional[torch.optim.Optimizer] = None,
        lr: float = 0.001,
        GEMS_INVERSE: bool = False,
        act_dtype: Optional[torch.dtype] = None,
        autocast_dtype: Optional[torch.dtype] = None,
        device: Optional[torch.device] = None,
        schedule: str = "gpipe",
        act_ckpt: bool = False,
    ):
        assert schedule in ("gpipe", "1f1b")
        self.schedule = schedule
        # activation checkpointing: recompute each cell's forward during
        # backward instead of storing activations. GPipe holds every
        # micro-batch's graph simultaneously, so this cuts peak activation
        # memory by ~the cell depth (the enabler for global batch 16 at
        # 2048^2 in 288 GB). The reference has no equivalent.
        self.act_ckpt = act_ckpt
        self.model_gen = model_gen
        self.comm = comm
        self.local_rank = local_rank
        self.split_size = model_gen.split_size
        self.split_rank = comm.get_split_rank(local_rank)
        self.batch_size = batch_size
        self.parts = parts
        assert batch_size % parts == 0, "batch_size must divide into parts"
        self.mb = batch_size // parts  # micro-batch size
        self.GEMS_INVERSE = GEMS_INVERSE
        self.device = device or current_device()
        self.autocast_dtype = autocast_dtype
        self.act_dtype = act_dtype or torch.float32

        self.models = model_gen.models  # local stage (after ready_model)
        assert self.models is not None, "call model_gen.ready_model first"
        self.criterion = criterion or nn.CrossEntropyLoss()
        self.optimizer = optimizer or torch.optim.SGD(
            self.models.parameters(), lr=lr, momentum=0.9
        )

        self.shape_list = model_gen.shape_list
        assert self.shape_list, "model_gen must have run get_output_shapes"

        self._init_peers()
        self._init_buffers()

        # side HIP stream for stage sends: the send's RCCL kernel waits
        # only on an event recorded when the activation is ready, not on
        # everything queued behind it on the compute stream — so a stage
        # send overlaps the NEXT micro-batch's compute.
        self._comm_stream = None
        if (
            self.device.type == "cuda"
            and torch.cuda.is_available()
            and os.environ.get("MPI4DL_P2P_SIDE_STREAM", "1") != "0"
        ):
            self._comm_stream = torch.cuda.Stream(device=self.device)

        # per-step state
        self.inputs: List = [None] * parts   # leaf inputs per part
        self.outputs: List = [None] * parts  # stage outputs per part
        self.loss_sum = 0.0
        self.correct_sum = 0
        self.seen = 0
        self._pending: List[p2p.Transfer] = []
        # False -> skip host-side loss/accuracy scalarisation (the
        # .item() syncs), making run_step hipGraph-capturable
        self.metrics_enabled = True

    # ------------------------------------------------------------------
    # topology
    # ------------------------------------------------------------------

    def _init_peers(self):
        """Prev/next stage peers (reference mp_pipeline.py:238-248)."""
        self.first_stage = self.split_rank == 0
        self.last_stage = self.split_rank == self.split_size - 1
        self.prev_rank = None
        self.next_rank = None
        if not self.first_stage:
            self.prev_rank = self.comm.engine_peer(self.local_rank - 1, self.GEMS_INVERSE)
        if not self.last_stage:
            self.next_rank = self.comm.engine_peer(self.local_rank + 1, self.GEMS_INVERSE)

    def _recv_shapes(self) -> List[tuple]:
        """Shapes of the activations this stage receives (= prev stage's
        outputs), with batch dim set to the micro-batch size."""
        spec = _as_list(self.shape_list[self.split_rank - 1])
        return [(self.mb,) + tuple(s[1:]) for s in spec]

    def _init_buffers(self):
        """Preallocated per-part recv buffers (mp_pipeline.py:250-292)."""
        self.input_buffers = None
        self.grad_buffers = None
        if not self.first_stage:
            self.input_buffers = [
                [
                    torch.zeros(s, device=self.device, dtype=self.act_dtype)
                    for s in self._recv_shapes()
                ]
                for _ in range(self.parts)
            ]
        if not self.last_stage:
            out_spec = _as_list(self.shape_list[self.split_rank])
            self.grad_buffers = [
                [
                    torch.zeros((self.mb,) + tuple(s[1:]), device=self.device, dtype=self.act_dtype)
                    for s in out_spec
                ]
                for _ in range(self.parts)
            ]

    # ------------------------------------------------------------------
    # forward / backward per micro-batch
    # ------------------------------------------------------------------

    def _run_stage(self, x):
        if self.autocast_dtype is not None and self.device.type == "cuda":
            with torch.autocast("cuda", dtype=self.autocast_dtype):
                return self._stage_cells(x)
        return self._stage_cells(x)

    def _stage_cells(self, x):
        if not (
            self.act_ckpt and self.models.training and torch.is_grad_enabled()
        ):
            return self.models(x)
        # per-cell non-reentrant checkpoint. Halo/P2P exchanges inside a
        # cell's forward re-run during recompute; tile ranks recompute the
        # same cell for the same micro-batch in the same backward slot, so
        # the exchanges pair up exactly as in the original forward (and
        # forward tags are disjoint from backward-grad tags).
        from torch.utils.checkpoint import checkpoint

        for cell in self.models:
            x = checkpoint(cell, x, use_reentrant=False)
        return x

    def _leaf(self, b):
        """Received buffer -> autograd leaf. Without autocast the compute
        dtype is fp32, so bf16 boundary messages are up-cast here (under
        autocast the model handles mixed dtypes itself)."""
        t = b.clone()
        if self.autocast_dtype is None and t.dtype not in (torch.float32,):
            t = t.float()
        return t.requires_grad_(True)

    def receive_input(self, part: int):
        bufs = self.input_buffers[part]
        with GLOBAL_TIMER.phase("pipeline/recv_act"):
            p2p.recv_tensors(bufs, self.prev_rank, tag_base=1000 + part * 16)
        leaves = [self._leaf(b) for b in bufs]
        return leaves[0] if len(leaves) == 1 else tuple(leaves)

    def _isend_side(self, ts, peer, tag_base):
        """Issue an isend from the comm stream, gated on an event that
        fires when the tensors are ready on the compute stream."""
        if self._comm_stream is None:
            return p2p.isend_tensors(ts, peer, tag_base=tag_base)
        ev = torch.cuda.Event()
        ev.record()
        with torch.cuda.stream(self._comm_stream):
            ev.wait()
            tr = p2p.isend_tensors(ts, peer, tag_base=tag_base)
            for t in ts:
                t.record_stream(self._comm_stream)
        return tr

    def send_output(self, y, part: int):
        ts = [t.to(self.act_dtype) for t in (y if isinstance(y, tuple) else (y,))]
        tr = self._isend_side(ts, self.next_rank, 1000 + part * 16)
        self._pending.append(tr)

    def forward_pass(self, data_x, data_y, part: int):
        """One micro-batch forward on this stage (mp_pipeline.py:434-473)."""
        if self.first_stage:
            x = data_x.to(self.device, non_blocking=True)
        else:
            x = self.receive_input(part)
        self.inputs[part] = x
        with GLOBAL_TIMER.phase("pipeline/forward"):
            y = self._run_stage(x)
        self.outputs[part] = y
        if self.last_stage:
            yl = data_y.to(self.device, non_blocking=True)
            logits = y[0] if isinstance(y, tuple) else y
            loss = self.criterion(logits.float(), yl)
            if self.metrics_enabled:
                self.loss_sum += float(loss.detach())
                with torch.no_grad():
                    self.correct_sum += int((logits.argmax(dim=1) == yl).sum())
                    self.seen += yl.numel()
            # keep scaled loss for backward
            self.outputs[part] = loss * (1.0 / self.parts)
        else:
            self.send_output(y, part)
        return self.outputs[part]

    def receive_output_grad(self, part: int):
        """Grad wrt this stage's outputs, from the next stage."""
        gbufs = self.grad_buffers[part]
        p2p.recv_tensors(gbufs, self.next_rank, tag_base=3000 + part * 16)
        return gbufs

    def send_input_grad(self, part: int):
        """Grad wrt this stage's received inputs, to the previous stage."""
        x = self.inputs[part]
        xs = list(x) if isinstance(x, tuple) else [x]
        gsend = [
            (t.grad if t.grad is not None else torch.zeros_like(t)).to(self.act_dtype)
            for t in xs
        ]
        tr = self._isend_side(gsend, self.prev_rank, 3000 + part * 16)
        self._pending.append(tr)

    def backward_pass(self, part: int):
        """One micro-batch backward (mp_pipeline.py:475-507)."""
        y = self.outputs[part]
        if self.last_stage:
            with GLOBAL_TIMER.phase("pipeline/backward"):
                y.backward()
        else:
            with GLOBAL_TIMER.phase("pipeline/recv_grad"):
                gbufs = self.receive_output_grad(part)
            ys = list(y) if isinstance(y, tuple) else [y]
            grads = [g.to(t.dtype) for g, t in zip(gbufs, ys)]
            with GLOBAL_TIMER.phase("pipeline/backward"):
                torch.autograd.backward(ys, grads)
        if not self.first_stage:
            self.send_input_grad(part)
        # free graph state for this part
        self.outputs[part] = None
        self.inputs[part] = None

    # ------------------------------------------------------------------
    # step / update
    # ------------------------------------------------------------------

    def run_step(self, inputs: Optional[torch.Tensor], labels: Optional[torch.Tensor]):
        """One training step over the whole batch (mp_pipeline.py:509-534):
        all forwards, then all backwards (GPipe fill-drain)."""
        self.loss_sum = 0.0
        self.correct_sum = 0
        self.seen = 0
        parts_x = [None] * self.parts
        parts_y = [None] * self.parts
        if inputs is not None and self.first_stage:
            parts_x = list(inputs.chunk(self.parts, dim=0))
        if labels is not None and self.last_stage:
            parts_y = list(labels.chunk(self.parts, dim=0))
        overlap = getattr(self.models, "_mpi4dl_overlap", None)

        def bwd(part):
            if overlap is not None:
                # only the LAST micro-batch's backward triggers the
                # bucketed allreduce (grads accumulate until then)
                overlap["sync_enabled"] = part == self.parts - 1
            self.backward_pass(part)

        if self.schedule == "1f1b" and self.split_size > 1:
            # PipeDream-flush: stage r admits (split_size - split_rank)
            # in-flight micro-batches, then strictly alternates 1F1B —
            # same math as GPipe (all grads before the update), but peak
            # live activations drop from `parts` to `stages - rank`.
            warm = min(self.parts, self.split_size - self.split_rank)
            fwd = bwd_i = 0
            for _ in range(warm):
                self.forward_pass(parts_x[fwd], parts_y[fwd], fwd)
                fwd += 1
            while fwd < self.parts:
                bwd(bwd_i)
                bwd_i += 1
                self.forward_pass(parts_x[fwd], parts_y[fwd], fwd)
                fwd += 1
            while bwd_i < self.parts:
                bwd(bwd_i)
                bwd_i += 1
        else:
            for part in range(self.parts):
                self.forward_pass(parts_x[part], parts_y[part], part)
            for part in range(self.parts):
                bwd(part)
        self._drain()
        return self.loss_sum / max(self.parts, 1), self.correct_sum, self.seen

    def _drain(self):
        for tr in self._pending:
            tr.wait()
        self._pending = []

    def update(self):
        """Optimizer step + grad reset (mp_pipeline.py:536). Grads may be
        views into a FlatGrads buffer — never set_to_none."""
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=False)

    @torch.no_grad()
    def run_eval(self, inputs, labels):
        """Forward-only pipeline pass (reference benchmarks' optional
        evaluation loop, e.g. benchmark_resnet_sp.py --enable-evaluation).
        Returns (mean loss, correct, seen) on the last stage."""
        was_training = self.models.training
        self.models.eval()
        self.loss_sum = 0.0
        self.correct_sum = 0
        self.seen = 0
        parts_x = [None] * self.parts
        parts_y = [None] * self.parts
        if inputs is not None and self.first_stage:
            parts_x = list(inputs.chunk(self.parts, dim=0))
        if labels is not None and self.last_stage:
            parts_y = list(labels.chunk(self.parts, dim=0))
        for part in range(self.parts):
            self.forward_pass(parts_x[part], parts_y[part], part)
            self.outputs[part] = None
            self.inputs[part] = None
        self._drain()
        if was_training:
            self.models.train()
        return self.loss_sum / max(self.parts, 1), self.correct_sum, self.seen
