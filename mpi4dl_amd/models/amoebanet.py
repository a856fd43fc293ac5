"""AmoebaNet-D (GPipe variant) — plain and spatially-parallel builders.

Reference parity: src/models/amoebanet.py (amoebanetd :535,
amoebanetd_spatial :618). The genotype (operation lists + concat
indices) is the published AmoebaNet-D architecture data used by GPipe /
the TensorFlow TPU implementation (see the reference's provenance note,
amoebanet.py:318-336); everything else here is our own implementation:

* spatial config flows through a SpatialPlan (ops/plan.py) instead of
  the reference's module-global dict (amoebanet.py:26-33);
* all BatchNorms are created through the plan so spatial cells get
  tile-synced statistics (ops/norm.py);
* 1x7/7x1 convs use per-axis halos ((0,3) / (3,0)) — the reference
  routes them through full square halos;
* avg_pool_3x3 keeps exact count_include_pad=False semantics in tile
  mode (the reference's spatial path silently changes them);
* the reference's max_pool_3x3 builder instantiates an AvgPool in BOTH
  its serial and spatial paths (amoebanet.py:108-125) — its published
  numbers ran avg pools there. Default here is the faithful genotype
  (real max pool); ``ref_quirks=True`` reproduces the reference's
  behaviour bitwise (tests/test_model_parity.py proves equality).

Cells pass ``(x, skip)`` tuples between stages — the pipeline and every
spatial seam handle multi-tensor activations.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from ..ops.plan import SpatialPlan
from ..ops.spatial_conv import HaloConv2d, HaloPool2d

Tensor = torch.Tensor

# ---------------------------------------------------------------------------
# Genotype (architecture data — AmoebaNet-D, GPipe/TF-TPU variant)
# ---------------------------------------------------------------------------

# (input_state_index, op_name) x 10 = 5 pairs; ops on input states (<2)
# use stride 2 in reduction cells.
NORMAL_OPERATIONS = [
    (1, "conv_1x1"),
    (1, "max_pool_3x3"),
    (1, "none"),
    (0, "conv_1x7_7x1"),
    (0, "conv_1x1"),
    (0, "conv_1x7_7x1"),
    (2, "max_pool_3x3"),
    (2, "none"),
    (1, "avg_pool_3x3"),
    (5, "conv_1x1"),
]
NORMAL_CONCAT = [0, 3, 4, 6]

REDUCTION_OPERATIONS = [
    (0, "max_pool_2x2"),
    (0, "max_pool_3x3"),
    (2, "none"),
    (1, "conv_3x3"),
    (2, "conv_1x7_7x1"),
    (2, "max_pool_3x3"),
    (3, "none"),
    (1, "max_pool_2x2"),
    (2, "avg_pool_3x3"),
    (3, "conv_1x1"),
]
REDUCTION_CONCAT = [4, 5, 6]


# ---------------------------------------------------------------------------
# Operations
# ---------------------------------------------------------------------------


def _conv(cin, cout, k, stride=1, padding=None, ctx=None, bias=False):
    from ..ops.conv_native import NativeConv2d

    if isinstance(k, int):
        k = (k, k)
    if padding is None:
        padding = ((k[0] - 1) // 2, (k[1] - 1) // 2)
    if ctx is None or (padding[0] == 0 and padding[1] == 0):
        return NativeConv2d(cin, cout, k, stride=stride, padding=padding, bias=bias)
    return HaloConv2d(cin, cout, k, stride=stride, padding=padding, bias=bias, **ctx)


class FactorizedReduce(nn.Module):
    """ReLU -> [1x1/2 conv || 1x1/2 conv] -> cat -> BN (reference :56-77,
    with the pixel-shift branch disabled exactly as the reference does)."""

    def __init__(self, cin, cout, mknorm=nn.BatchNorm2d):
        super().__init__()
        from ..ops.conv_native import NativeConv2d

        self.relu = nn.ReLU(inplace=False)
        self.conv1 = NativeConv2d(cin, cout // 2, 1, stride=2, bias=False)
        self.conv2 = NativeConv2d(cin, cout // 2, 1, stride=2, bias=False)
        self.bn = mknorm(cout)

    def forward(self, x):
        x = self.relu(x)
        x = torch.cat([self.conv1(x), self.conv2(x)], dim=1)
        return self.bn(x)


def relu_conv_bn(cin, cout, mknorm=nn.BatchNorm2d):
    from ..ops.conv_native import NativeConv2d

    return nn.Sequential(
        nn.ReLU(inplace=False),
        NativeConv2d(cin, cout, 1, stride=1, bias=False),
        mknorm(cout),
    )


def _bn_relu(mknorm, ch):
    """BN with fused ReLU (TileBatchNorm2d relu=True); an Identity keeps
    the Sequential indices stable where the ReLU module used to sit."""
    bn = mknorm(ch)
    if hasattr(bn, "relu"):
        bn.relu = True
        return bn
    return nn.Sequential(bn, nn.ReLU(inplace=False))


def make_op(name: str, c: int, stride: int, ctx, mknorm,
            ref_quirks: bool = False) -> nn.Module:
    if name == "none":
        return (
            nn.Identity() if stride == 1 else FactorizedReduce(c, c, mknorm)
        )
    if name == "avg_pool_3x3":
        # plain path also routes through HaloPool2d (num_spatial_parts=1)
        # so the gemscore pool kernels run on GPU
        return HaloPool2d(
            "avg", 3, stride=stride, padding=1, count_include_pad=False,
            **(ctx or {})
        )
    if name == "max_pool_3x3":
        if ref_quirks:
            # the reference's max_pool_3x3 builder instantiates an AVG
            # pool in both its serial and spatial paths
            # (amoebanet.py:108-125) — its published numbers ran this.
            # Default (faithful genotype) uses a real max pool.
            return HaloPool2d(
                "avg", 3, stride=stride, padding=1,
                count_include_pad=False, **(ctx or {})
            )
        return HaloPool2d("max", 3, stride=stride, padding=1, **(ctx or {}))
    if name == "max_pool_2x2":
        # padding 0, stride 2 in reduction cells: tile-local, no halo
        return HaloPool2d("max", 2, stride=stride, padding=0)
    # conv ops START with ReLU of their input state in the reference; the
    # Cell precomputes relu(state) ONCE and passes it to conv ops (pool /
    # none ops get the raw state), removing ~6 redundant full-tensor ReLU
    # kernels per cell. The leading ReLU module is therefore dropped here.
    if name == "conv_1x1":
        from ..ops.conv_native import NativeConv2d

        return nn.Sequential(
            NativeConv2d(c, c, 1, stride=stride, bias=False),
            mknorm(c),
        )
    if name == "conv_3x3":
        from ..ops.conv_native import NativeConv2d

        return nn.Sequential(
            NativeConv2d(c, c // 4, 1, bias=False),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            _conv(c // 4, c // 4, 3, stride=stride, ctx=ctx),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            NativeConv2d(c // 4, c, 1, bias=False),
            mknorm(c),
        )
    if name == "conv_1x7_7x1":
        from ..ops.conv_native import NativeConv2d

        return nn.Sequential(
            NativeConv2d(c, c // 4, 1, stride=1, bias=False),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            _conv(c // 4, c // 4, (1, 7), stride=(1, stride), padding=(0, 3), ctx=ctx),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            _conv(c // 4, c // 4, (7, 1), stride=(stride, 1), padding=(3, 0), ctx=ctx),
            _bn_relu(mknorm, c // 4),
            nn.Identity(),
            NativeConv2d(c // 4, c, 1, stride=1, bias=False),
            mknorm(c),
        )
    raise ValueError(f"unknown op {name}")


# ---------------------------------------------------------------------------
# Cells
# ---------------------------------------------------------------------------


class Stem(nn.Module):
    """ReLU -> 3x3/2 conv -> BN (reference Stem :417-447 order kept)."""

    def __init__(self, channels: int, ctx=None, mknorm=nn.BatchNorm2d):
        super().__init__()
        self.relu = nn.ReLU(inplace=False)
        self.conv = _conv(3, channels, 3, stride=2, padding=(1, 1), ctx=ctx)
        self.bn = mknorm(channels)

    def forward(self, x):
        return self.bn(self.conv(self.relu(x)))


class Cell(nn.Module):
    """One AmoebaNet cell (reference Cell :449-533): takes (s1, s2) — or a
    single tensor used for both — reduces them to ``channels``, applies the
    5 genotype pairs, concats the selected states; returns (out, skip=s1_in).
    """

    def __init__(
        self,
        channels_prev_prev: int,
        channels_prev: int,
        channels: int,
        reduction: bool,
        reduction_prev: bool,
        ctx=None,
        mknorm=nn.BatchNorm2d,
        ref_quirks: bool = False,
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
        # sum states consumed ONLY by the final concat never materialise:
        # AddCat writes h1+h2 straight into their channel slice (ops/fuse.py)
        self._cat_only = frozenset(
            i for i in self.concat if i >= 2 and i not in set(self.indices)
        )
        self.operations = nn.ModuleList()
        for i, name in ops:
            stride = 2 if (reduction and i < 2) else 1
            self.operations.append(
                make_op(name, channels, stride, ctx, mknorm,
                        ref_quirks=ref_quirks)
            )

    def forward(self, input_or_states):
        if isinstance(input_or_states, tuple):
            s1, s2 = input_or_states
        else:
            s1 = s2 = input_or_states
        skip = s1
        states = [self.reduce1(s1), self.reduce2(s2)]
        relu_cache = {}

        def get(idx, want_relu):
            if not want_relu:
                return states[idx]
            if idx not in relu_cache:
                relu_cache[idx] = torch.relu(states[idx])
            return relu_cache[idx]

        pending = {}
        for i in range(0, len(self.operations), 2):
            h1 = self.operations[i](get(self.indices[i], self.wants_relu[i]))
            h2 = self.operations[i + 1](
                get(self.indices[i + 1], self.wants_relu[i + 1])
            )
            sidx = 2 + i // 2
            if sidx in self._cat_only:
                pending[sidx] = (h1, h2)
                states.append(None)
            else:
                states.append(h1 + h2)
        from ..ops.fuse import add_cat

        entries = [
            pending[i] if i in pending else (states[i], None)
            for i in self.concat
        ]
        return add_cat(entries), skip


class Classify(nn.Module):
    def __init__(self, channels_prev: int, num_classes: int):
        super().__init__()
        self.pool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(channels_prev, num_classes)

    def forward(self, states):
        x = states[0] if isinstance(states, tuple) else states
        return self.fc(torch.flatten(self.pool(x), 1))


# ---------------------------------------------------------------------------
# Builder
# ---------------------------------------------------------------------------


def amoebanetd(
    num_classes: int = 10,
    num_layers: int = 6,
    num_filters: int = 64,
    plan: Optional[SpatialPlan] = None,
    ref_quirks: bool = False,
) -> nn.Sequential:
    """AmoebaNet-D as a flat Sequential of cells (reference :535-616).
    ``plan`` != None builds the spatial variant (reference
    amoebanetd_spatial :618-737) — halo ops + tile-synced BN on the
    cells assigned to spatial partitions."""
    assert num_layers % 3 == 0, "num_layers must be divisible by 3"
    repeat = num_layers // 3

    channels = num_filters // 4
    state = {
        "c_pp": channels,
        "c_p": channels,
        "c": channels,
        "red_prev": False,
    }
    cells: List[nn.Module] = []

    def ctx():
        return plan.ctx(len(cells)) if plan is not None else None

    def mknorm():
        if plan is None:
            # TileBatchNorm2d(group=None) == BatchNorm2d numerically but
            # takes the gemscore fused path on GPU
            from ..ops.norm import TileBatchNorm2d

            return TileBatchNorm2d
        i = len(cells)
        return lambda ch: plan.norm(ch, i)

    def add_cell(reduction: bool, scale: int):
        state["c"] *= scale
        cell = Cell(
            state["c_pp"],
            state["c_p"],
            state["c"],
            reduction,
            state["red_prev"],
            ctx=ctx(),
            mknorm=mknorm(),
            ref_quirks=ref_quirks,
        )
        state["c_pp"] = state["c_p"]
        state["c_p"] = state["c"] * len(cell.concat)
        state["red_prev"] = reduction
        cells.append(cell)

    cells.append(Stem(channels, ctx=ctx(), mknorm=mknorm()))
    add_cell(True, 2)   # stem2
    add_cell(True, 2)   # stem3
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


def amoebanetd_spatial(
    plan: SpatialPlan,
    num_classes: int = 10,
    num_layers: int = 6,
    num_filters: int = 64,
) -> nn.Sequential:
    """torchgems-compat entry point (reference amoebanet.py:618)."""
    return amoebanetd(num_classes, num_layers, num_filters, plan=plan)
