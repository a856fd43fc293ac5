"""Spatial ("D1") ResNet builders: cells on spatial partitions use
halo-aware convs/pools and tile-synced BatchNorm, later cells are plain.

Reference parity: src/models/resnet_spatial.py (get_resnet_v1 :299,
get_resnet_v2 :545). Cell indices line up exactly with models/resnet.py
builders so the same balance/partitioning applies to both. Improvement
over the reference: BatchNorm inside spatial partitions syncs its batch
statistics across the tile group (ops/norm.py), so SP training computes
exactly the single-GPU function (the reference's per-tile BN does not).
"""

from __future__ import annotations

from typing import Optional

import torch.nn as nn

from ..ops.plan import SpatialPlan
from ..ops.conv_native import NativeConv2d
from ..ops.spatial_conv import HaloConv2d, HaloPool2d
from .resnet import Head  # noqa: F401


def _bn_relu(mknorm, ch):
    bn = mknorm(ch)
    if hasattr(bn, "relu"):
        bn.relu = True
        return bn
    return nn.Sequential(bn, nn.ReLU(inplace=False))


def sconv(in_ch, out_ch, k, stride, ctx: Optional[dict], bias=False):
    from ..ops.conv_native import NativeConv2d

    if ctx is None:
        return NativeConv2d(in_ch, out_ch, k, stride=stride, padding=k // 2, bias=bias)
    return HaloConv2d(in_ch, out_ch, k, stride=stride, bias=bias, **ctx)


class BasicBlockV1S(nn.Module):
    """Spatial variant of BasicBlockV1 (reference make_cell_v1_spatial :183)."""

    def __init__(self, in_ch, out_ch, stride, ctx, mknorm=nn.BatchNorm2d):
        super().__init__()
        self.body = nn.Sequential(
            sconv(in_ch, out_ch, 3, stride, ctx, bias=True),
            _bn_relu(mknorm, out_ch),
            nn.Identity(),
            sconv(out_ch, out_ch, 3, 1, ctx, bias=True),
            mknorm(out_ch),
        )
        self.proj = None
        if stride != 1 or in_ch != out_ch:
            # 1x1 stride-s conv needs no halo
            self.proj = NativeConv2d(in_ch, out_ch, 1, stride=stride, bias=True)
        self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        s = x if self.proj is None else self.proj(x)
        return self.act(self.body(x) + s)


class BottleneckV2S(nn.Module):
    """Spatial variant of BottleneckV2 (reference make_cell_v2_spatial
    :375): pre-act 3x3(s) -> 3x3 -> 1x1; both 3x3s are halo convs, the
    1x1s and the raw-input projection need no halo."""

    def __init__(self, in_ch, mid_ch, out_ch, stride, ctx,
                 mknorm=nn.BatchNorm2d, preact=True):
        super().__init__()
        self.pre1 = (
            nn.Sequential(_bn_relu(mknorm, in_ch), nn.Identity())
            if preact
            else nn.Identity()
        )
        self.conv1 = sconv(in_ch, mid_ch, 3, stride, ctx, bias=True)
        self.pre2 = nn.Sequential(_bn_relu(mknorm, mid_ch), nn.Identity())
        self.conv2 = sconv(mid_ch, mid_ch, 3, 1, ctx, bias=True)
        self.pre3 = nn.Sequential(_bn_relu(mknorm, mid_ch), nn.Identity())
        self.conv3 = nn.Conv2d(mid_ch, out_ch, 1)
        self.proj = None
        if stride != 1 or in_ch != out_ch:
            self.proj = nn.Conv2d(in_ch, out_ch, 1, stride=stride)

    def forward(self, x):
        y = self.conv1(self.pre1(x))
        y = self.conv2(self.pre2(y))
        y = self.conv3(self.pre3(y))
        s = x if self.proj is None else self.proj(x)
        return s + y


class StemS(nn.Module):
    """Spatial stem: 7x7/2 halo conv + BN/ReLU + 3x3/2 halo max pool
    (or the reference's stride-1 3x3 stem with ref_stem=True)."""

    def __init__(self, in_ch, filters, image_size, ctx,
                 mknorm=nn.BatchNorm2d, ref_stem=False):
        super().__init__()
        if image_size >= 128 and not ref_stem:
            self.ops = nn.Sequential(
                sconv(in_ch, filters, 7, 2, ctx),
                _bn_relu(mknorm, filters),
                nn.Identity(),
                HaloPool2d("max", 3, stride=2, padding=1, **(ctx or {})),
            )
        else:
            self.ops = nn.Sequential(
                sconv(in_ch, filters, 3, 1, ctx, bias=True),
                mknorm(filters),
                nn.ReLU(inplace=True),
            )

    def forward(self, x):
        return self.ops(x)


def _build(input_shape, num_classes, n, num_filters, plan, version,
           ref_stem=False):
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
    for group in range(3):
        if version == 1:
            out_ch = num_filters * (2**group)
            for block in range(n):
                stride = 2 if (group > 0 and block == 0) else 1
                cells.append(BasicBlockV1S(ch, out_ch, stride, ctx(), mknorm()))
                ch = out_ch
        else:
            # reference width schedule: stage 0 4x / stride 1, later 2x
            mid = num_filters if group == 0 else ch
            out = mid * (4 if group == 0 else 2)
            for block in range(n):
                stride = 2 if (group > 0 and block == 0) else 1
                cells.append(BottleneckV2S(
                    ch, mid, out, stride, ctx(), mknorm(),
                    preact=not (group == 0 and block == 0),
                ))
                ch = out
    cells.append(Head(ch, num_classes, final_bn=(version == 2), mknorm=mknorm()))
    return nn.Sequential(*cells)


def get_resnet_v1(
    input_shape, num_classes=10, n=3, num_filters=16,
    plan: Optional[SpatialPlan] = None, ref_stem=False,
):
    return _build(input_shape, num_classes, n, num_filters, plan, version=1,
                  ref_stem=ref_stem)


def get_resnet_v2(
    input_shape, num_classes=10, n=12, num_filters=16,
    plan: Optional[SpatialPlan] = None, ref_stem=False,
):
    return _build(input_shape, num_classes, n, num_filters, plan, version=2,
                  ref_stem=ref_stem)


def get_resnet101_cells(
    input_shape, num_classes=1000, width=64, plan: Optional[SpatialPlan] = None
):
    """Spatial ResNet-101 (BASELINE config 4). Cell layout matches
    models/resnet.py:get_resnet101_cells."""
    _, in_ch, H, W = input_shape
    cells = []

    def ctx():
        return plan.ctx(len(cells)) if plan is not None else None

    def mknorm():
        if plan is None:
            return nn.BatchNorm2d
        i = len(cells)
        return lambda ch: plan.norm(ch, i)

    cells.append(StemS(in_ch, width, min(H, W), ctx(), mknorm()))
    ch = width
    for group, blocks in enumerate([3, 4, 23, 3]):
        mid = width * (2**group)
        for b in range(blocks):
            stride = 2 if (group > 0 and b == 0) else 1
            cells.append(BottleneckV2S(ch, mid, mid * 4, stride, ctx(), mknorm()))
            ch = mid * 4
    cells.append(Head(ch, num_classes, final_bn=True, mknorm=mknorm()))
    return nn.Sequential(*cells)
