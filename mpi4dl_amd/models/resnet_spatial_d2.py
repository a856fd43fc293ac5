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
    """Pre-activation bottleneck consuming 1 surplus pixel (stride 1).

    ``pre_h`` > 0 prepends the fused halo exchange that grants the next
    ``pre_h`` blocks their surplus.
    """

    expansion = 4

    def __init__(self, in_ch, mid_ch, ctx, mknorm=nn.BatchNorm2d, pre_h: int = 0):
        super().__init__()
        out_ch = mid_ch * self.expansion
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
        self.pre = nn.Sequential(mknorm(in_ch), nn.ReLU(inplace=True))
        self.conv1 = nn.Conv2d(in_ch, mid_ch, 1, bias=False)
        self.bn1 = mknorm(mid_ch)
        self.act = nn.ReLU(inplace=True)
        # the 3x3: interior sides unpadded (consume surplus), boundary
        # sides freshly zero-padded
        self.conv2 = nn.Conv2d(mid_ch, mid_ch, 3, stride=1, padding=0, bias=False)
        self.bn2 = mknorm(mid_ch)
        self.conv3 = nn.Conv2d(mid_ch, out_ch, 1, bias=False)
        self.proj = None
        if in_ch != out_ch:
            self.proj = nn.Conv2d(in_ch, out_ch, 1, bias=False)
        self.crop = CropInterior(self.exchanger, 1)

    def forward(self, x):
        if self.pre_h > 0:
            x = halo_pad_d2(x, self.pre_h, self.exchanger, self.grad_mode)
        h = self.pre(x)
        s = x if self.proj is None else self.proj(h)
        y = self.act(self.bn1(self.conv1(h)))
        y = outer_pad_only(y, self.layout, self.tile, 1) if self.exchanger is not None else nn.functional.pad(y, (1, 1, 1, 1))
        y = self.conv3(self.act(self.bn2(self.conv2(y))))
        return y + self.crop(s)


def get_resnet_v2(
    input_shape,
    num_classes: int = 10,
    n: int = 12,
    num_filters: int = 16,
    plan: Optional[SpatialPlan] = None,
    fused_layers: int = 4,
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

    cells.append(StemS(in_ch, num_filters, min(H, W), ctx(), mknorm()))
    ch = num_filters
    for group in range(3):
        mid = num_filters * (2**group)
        surplus_left = 0
        for block in range(n):
            stride = 2 if (group > 0 and block == 0) else 1
            c = ctx()
            out_ch = mid * BottleneckV2D2.expansion
            d1_style = (
                c is None
                or stride != 1
                or block == 0  # channel-change block: keep D1 (has proj)
            )
            if d1_style:
                cells.append(BottleneckV2S(ch, mid, stride, c, mknorm()))
                surplus_left = 0
            else:
                pre_h = 0
                if surplus_left == 0:
                    pre_h = min(fused_layers, n - block)
                    surplus_left = pre_h
                cells.append(
                    BottleneckV2D2(ch, mid, c, mknorm(), pre_h=pre_h)
                )
                surplus_left -= 1
            ch = out_ch
    cells.append(Head(ch, num_classes, final_bn=True, mknorm=mknorm()))
    return nn.Sequential(*cells)
