"""ResNet v2 "Design 2" (fused halo) builder.

Reference parity: src/models/resnet_spatial_d2.py — one LARGE halo
exchange every ``fused_layers`` blocks instead of one small exchange per
conv (get_resnet_v2 :578, halo_len arithmetic :651-697, residual crop
:462-469). The exchange gives each tile a *surplus* ring of
``fused_layers`` correct neighbour pixels on its interior sides; each
3x3 conv then runs unpadded on interior sides (consuming 1 surplus) and
freshly zero-padded on image-boundary sides (spatial.py:67-111), so
boundary behaviour matches the undistributed conv exactly.

Design differences from the reference:
* the fused exchange is folded INTO the first cell of each fused chunk,
  so the D2 model has the SAME cell count/indices as the D1/plain
  builders — no separate returned balance (the reference must return
  (model, balance) because its halo layers are extra cells, :726);
* stride-2 / projection blocks run D1-style (their own 1-pixel halo
  exchange at surplus 0) instead of the reference's stride-aware
  halo_len formulas — simpler and exchange-count-equivalent;
* with BatchNorm in eval mode the D2 forward is bitwise the serial
  model; in training the only deviation is tile-sync BN statistics
  counting the surplus pixels (the reference's per-tile BN deviates far
  more).
"""

from __future__ import annotations

from typing import Optional

import torch.nn as nn

from ..ops.halo import halo_pad_d2
from ..ops.plan import SpatialPlan
from ..ops.spatial_conv import outer_pad_only
from .resnet import Head
from .resnet_spatial import BottleneckV2S, StemS


class CropInterior(nn.Module):
    """Crop n pixels from each interior side (surplus consumption for the
    residual/skip path — reference resnet_spatial_d2.py:462-469)."""

    def __init__(self, exchanger, n: int):
        super().__init__()
        self.exchanger = exchanger
        self.n = n

    def forward(self, x):
        if self.exchanger is None or self.n == 0:
            return x
        t, b, l, r = self.exchanger.pads_d2(self.n)
        H, W = x.shape[-2], x.shape[-1]
        return x[:, :, t : H - b, l : W - r]


class BottleneckV2D2(nn.Module):
    """Pre-activation 3x3 -> 3x3 -> 1x1 block (reference make_cell_v2
    structure) consuming TWO surplus pixels per block (one per 3x3;
    stride-1 / no-projection blocks only — others run D1-style).

    ``pre_h`` > 0 prepends the fused halo exchange granting ``pre_h``
    surplus pixels to the blocks that follow.
    """

    def __init__(self, in_ch, mid_ch, out_ch, ctx, mknorm=nn.BatchNorm2d,
                 pre_h: int = 0):
        super().__init__()
        assert in_ch == out_ch, "D2 blocks are the no-projection blocks"
        from ..ops.halo import HaloExchanger, TileLayout

        self.layout = TileLayout(ctx["num_spatial_parts"], ctx["slice_method"])
        self.tile = ctx["spatial_local_rank"]
        self.exchanger = (
            HaloExchanger(self.layout, self.tile, ctx["rank_of_tile"])
            if ctx["num_spatial_parts"] > 1
            else None
        )
        self.grad_mode = ctx.get("grad_mode", "exact")
        self.pre_h = pre_h
        self.pre1 = nn.Sequential(mknorm(in_ch), nn.ReLU(inplace=True))
        # 3x3s: interior sides unpadded (consume surplus), image-boundary
        # sides freshly zero-padded (outer_pad_only)
        self.conv1 = nn.Conv2d(in_ch, mid_ch, 3, padding=0)
        self.pre2 = nn.Sequential(mknorm(mid_ch), nn.ReLU(inplace=True))
        self.conv2 = nn.Conv2d(mid_ch, mid_ch, 3, padding=0)
        self.pre3 = nn.Sequential(mknorm(mid_ch), nn.ReLU(inplace=True))
        self.conv3 = nn.Conv2d(mid_ch, out_ch, 1)
        self.crop = CropInterior(self.exchanger, 2)

    def _outer_pad(self, y):
        if self.exchanger is not None:
            return outer_pad_only(y, self.layout, self.tile, 1)
        return nn.functional.pad(y, (1, 1, 1, 1))

    def forward(self, x):
        if self.pre_h > 0:
            x = halo_pad_d2(x, self.pre_h, self.exchanger, self.grad_mode)
        y = self.conv1(self._outer_pad(self.pre1(x)))
        y = self.conv2(self._outer_pad(self.pre2(y)))
        y = self.conv3(self.pre3(y))
        return y + self.crop(x)


def get_resnet_v2(
    input_shape,
    num_classes: int = 10,
    n: int = 12,
    num_filters: int = 16,
    plan: Optional[SpatialPlan] = None,
    fused_layers: int = 4,
    ref_stem: bool = False,
):
    """D2 ResNet v2: cell layout identical to resnet/resnet_spatial v2
    builders (stem + 3n bottlenecks + head); spatial cells inside
    stride-1 runs use the fused-halo design."""
    _, in_ch, H, W = input_shape
    cells = []

    def ctx():
        return plan.ctx(len(cells)) if plan is not None else None

    def mknorm():
        if plan is None:
            return nn.BatchNorm2d
        i = len(cells)
        return lambda ch: plan.norm(ch, i)

    cells.append(StemS(in_ch, num_filters, min(H, W), ctx(), mknorm(),
                       ref_stem=ref_stem))
    ch = num_filters
    mid = num_filters
    for group in range(3):
        # reference width schedule: stage 0 expands 4x (stride 1), later
        # stages 2x with stride 2 at block 0 (resnet.py:288-300)
        out_ch = mid * (4 if group == 0 else 2)
        surplus_left = 0
        for block in range(n):
            stride = 2 if (group > 0 and block == 0) else 1
            c = ctx()
            d1_style = (
                c is None
                or stride != 1
                or block == 0  # channel-change block: keep D1 (has proj)
            )
            if d1_style:
                cells.append(BottleneckV2S(
                    ch, mid, out_ch, stride, c, mknorm(),
                    preact=not (group == 0 and block == 0),
                ))
                surplus_left = 0
            else:
                pre_h = 0
                if surplus_left == 0:
                    # 2 surplus pixels per fused block (two 3x3 convs)
                    nblk = min(fused_layers, n - block)
                    pre_h = 2 * nblk
                    surplus_left = nblk
                cells.append(
                    BottleneckV2D2(ch, mid, out_ch, c, mknorm(), pre_h=pre_h)
                )
                surplus_left -= 1
            ch = out_ch
        mid = out_ch
    cells.append(Head(ch, num_classes, final_bn=True, mknorm=mknorm()))
    return nn.Sequential(*cells)
