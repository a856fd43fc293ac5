"""AmoebaNet-D "Design 2" (fused halo) — one cell-level halo exchange
feeds all conv ops that read the same state.

Reference parity: src/models/amoebanet_d2.py (Cell_D2 :569-676 — it
pre-halos s3 = halo(s1, 3), s4 = halo(s2, 2), crops s5, and rewires the
op input indices; the _d2 op variants run with padding=0 :88-311).

Our design keeps the same economics with exact semantics:
* for each input state consumed by stride-1 conv ops, ONE interior-side
  halo_pad_d2 with the max halo those ops need (3 when any 1x7/7x1 op
  reads it, else 1) replaces the per-op exchanges — e.g. the two
  conv_1x7_7x1 ops on state 0 of a normal cell share one exchange
  (4 messages -> 1);
* conv ops consume the surplus via outer-pad-only convs (HaloConv2d
  d2=True), so image-boundary behaviour stays exactly the serial conv;
* pools and stride-2 ops keep their D1 per-op exchange (pool divisor
  geometry under surplus is not worth the complexity — the reference's
  D2 pools silently change semantics instead);
* the only train-mode deviation from D1 is BN statistics inside
  conv_1x7_7x1 seeing surplus pixels (eval mode is exact).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..ops.halo import HaloExchanger, TileLayout, halo_pad_d2
from ..ops.plan import SpatialPlan
from .amoebanet import (
    _bn_relu,
    NORMAL_CONCAT,
    NORMAL_OPERATIONS,
    REDUCTION_CONCAT,
    REDUCTION_OPERATIONS,
    Classify,
    FactorizedReduce,
    Stem,
    make_op,
    relu_conv_bn,
)
from .amoebanet import _conv  # noqa: F401  (re-exported for parity tools)


def _op_halo_need(name: str) -> int:
    """Interior halo a stride-1 conv op consumes (0 = no exchange)."""
    if name == "conv_1x7_7x1":
        return 3
    if name == "conv_3x3":
        return 1
    return 0


def _make_op_d2(name, c, stride, ctx, mknorm):
    """conv ops in d2 mode (outer pads only); everything else D1."""
    if ctx is not None and stride == 1 and _op_halo_need(name):
        d2ctx = dict(ctx)
        if name == "conv_3x3":
            return nn.Sequential(
                nn.Conv2d(c, c // 4, 1, bias=False),
                _bn_relu(mknorm, c // 4),
                nn.Identity(),
                _d2conv(c // 4, c // 4, (3, 3), (1, 1), d2ctx),
                _bn_relu(mknorm, c // 4),
                nn.Identity(),
                nn.Conv2d(c // 4, c, 1, bias=False),
                mknorm(c),
            )
        # conv_1x7_7x1
        return nn.Sequential(
            nn.Conv2d(c, c // 4, 1, stride=1, bias=False),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            _d2conv(c // 4, c // 4, (1, 7), (0, 3), d2ctx),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            _d2conv(c // 4, c // 4, (7, 1), (3, 0), d2ctx),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            nn.Conv2d(c // 4, c, 1, stride=1, bias=False),
            mknorm(c),
        )
    return make_op(name, c, stride, ctx, mknorm)


def _d2conv(cin, cout, k, pad, ctx):
    from ..ops.spatial_conv import HaloConv2d

    # halo_len=(0,0): the CELL already exchanged the surplus; the conv
    # only applies fresh zero pads on image-boundary sides (outer_pad)
    return HaloConv2d(
        cin, cout, k, stride=1, padding=pad, bias=False, d2=True,
        halo_len=(0, 0), **ctx
    )


class CellD2(nn.Module):
    """Cell with per-state fused halo exchange for its conv ops.

    Parameter layout matches amoebanet.Cell exactly (same ops in the
    same order; nn.ReLU leading modules were already factored out by the
    base implementation)."""

    def __init__(
        self,
        channels_prev_prev,
        channels_prev,
        channels,
        reduction,
        reduction_prev,
        ctx=None,
        mknorm=nn.BatchNorm2d,
    ):
        super().__init__()
        self.reduce1 = relu_conv_bn(channels_prev, channels, mknorm)
        self.reduce2: nn.Module = nn.Identity()
        if reduction_prev:
            self.reduce2 = FactorizedReduce(channels_prev_prev, channels, mknorm)
        elif channels_prev_prev != channels:
            self.reduce2 = relu_conv_bn(channels_prev_prev, channels, mknorm)

        ops = REDUCTION_OPERATIONS if reduction else NORMAL_OPERATIONS
        self.concat = REDUCTION_CONCAT if reduction else NORMAL_CONCAT
        self.indices = [i for i, _ in ops]
        self.wants_relu = [name.startswith("conv") for _, name in ops]
        self.op_strides = [2 if (reduction and i < 2) else 1 for i, _ in ops]
        # halo need per op (fused exchange only for stride-1 convs)
        self.op_need = [
            _op_halo_need(name) if (st == 1 and ctx is not None) else 0
            for (_, name), st in zip(ops, self.op_strides)
        ]
        self.operations = nn.ModuleList()
        for (i, name), st in zip(ops, self.op_strides):
            self.operations.append(_make_op_d2(name, channels, st, ctx, mknorm))

        self.grad_mode = (ctx or {}).get("grad_mode", "exact")
        if ctx is not None and ctx["num_spatial_parts"] > 1:
            layout = TileLayout(ctx["num_spatial_parts"], ctx["slice_method"])
            self.exchanger = HaloExchanger(
                layout, ctx["spatial_local_rank"], ctx["rank_of_tile"]
            )
        else:
            self.exchanger = None

    def forward(self, input_or_states):
        if isinstance(input_or_states, tuple):
            s1, s2 = input_or_states
        else:
            s1 = s2 = input_or_states
        skip = s1
        states = [self.reduce1(s1), self.reduce2(s2)]
        relu_cache = {}
        halo_cache = {}  # (state idx, h) -> exchanged tensor

        def get(pos):
            idx = self.indices[pos]
            t = states[idx]
            if self.wants_relu[pos]:
                if idx not in relu_cache:
                    relu_cache[idx] = torch.relu(t)
                t = relu_cache[idx]
            need = self.op_need[pos]
            if need and self.exchanger is not None:
                # fused exchange: max need over the cell's ops for this
                # state, computed lazily and shared
                key = idx
                if key not in halo_cache:
                    max_h = max(
                        self.op_need[p]
                        for p in range(len(self.op_need))
                        if self.indices[p] == idx and self.op_need[p]
                    )
                    halo_cache[key] = (
                        halo_pad_d2(t, max_h, self.exchanger, self.grad_mode),
                        max_h,
                    )
                et, have = halo_cache[key]
                if have > need:
                    tb, bb, lb, rb = self.exchanger.pads_d2(have - need)
                    H, W = et.shape[-2], et.shape[-1]
                    et = et[:, :, tb : H - bb, lb : W - rb]
                t = et
            return t

        used = set(self.indices)
        pending = {}
        for i in range(0, len(self.operations), 2):
            h1 = self.operations[i](get(i))
            h2 = self.operations[i + 1](get(i + 1))
            sidx = 2 + i // 2
            if sidx in self.concat and sidx not in used:
                pending[sidx] = (h1, h2)
                states.append(None)
            else:
                states.append(h1 + h2)
        from ..ops.fuse import add_cat

        return add_cat([
            pending[i] if i in pending else (states[i], None)
            for i in self.concat
        ]), skip


def amoebanetd_d2(
    num_classes: int = 10,
    num_layers: int = 6,
    num_filters: int = 64,
    plan: Optional[SpatialPlan] = None,
) -> nn.Sequential:
    """D2 AmoebaNet-D; cell count/indices match amoebanet.amoebanetd."""
    assert num_layers % 3 == 0
    repeat = num_layers // 3
    channels = num_filters // 4
    state = {"c_pp": channels, "c_p": channels, "c": channels, "red_prev": False}
    cells = []

    def ctx():
        return plan.ctx(len(cells)) if plan is not None else None

    def mknorm():
        if plan is None:
            from ..ops.norm import TileBatchNorm2d

            return TileBatchNorm2d
        i = len(cells)
        return lambda ch: plan.norm(ch, i)

    def add_cell(reduction, scale):
        state["c"] *= scale
        cell = CellD2(
            state["c_pp"], state["c_p"], state["c"], reduction,
            state["red_prev"], ctx=ctx(), mknorm=mknorm(),
        )
        state["c_pp"] = state["c_p"]
        state["c_p"] = state["c"] * len(cell.concat)
        state["red_prev"] = reduction
        cells.append(cell)

    cells.append(Stem(channels, ctx=ctx(), mknorm=mknorm()))
    add_cell(True, 2)
    add_cell(True, 2)
    for _ in range(repeat):
        add_cell(False, 1)
    add_cell(True, 2)
    for _ in range(repeat):
        add_cell(False, 1)
    add_cell(True, 2)
    for _ in range(repeat):
        add_cell(False, 1)
    cells.append(Classify(state["c_p"], num_classes))
    return nn.Sequential(*cells)
