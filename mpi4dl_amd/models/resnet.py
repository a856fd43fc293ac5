"""ResNet v1 (6n+2) and v2 (9n+2, pre-activation bottleneck) builders.

Reference parity: src/models/resnet.py (get_resnet_v1 :145,
get_resnet_v2 :270). Built as a flat ``nn.Sequential`` of coarse cells
(one residual block per cell) so the pipeline partitioner can split at
cell granularity. Residual connections live *inside* a cell, so every
inter-cell activation is a single tensor — the simple pipeline case.

Deliberate deviations from the reference:
* no softmax before CrossEntropyLoss (reference double-softmaxes,
  resnet.py:140,265);
* the classifier head is one cell (pool + flatten + linear).

``device='meta'`` builds the model with meta parameters — used for
shape inference and for ranks that never materialise remote stages.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..ops.conv_native import NativeConv2d


def conv_bn_act(
    in_ch: int,
    out_ch: int,
    kernel: int = 3,
    stride: int = 1,
    act: bool = True,
    bn: bool = True,
):
    # bias=True even under BN: the reference's resnet_layer uses the
    # nn.Conv2d default everywhere (resnet.py:40-46)
    pad = kernel // 2
    layers = [NativeConv2d(in_ch, out_ch, kernel, stride=stride, padding=pad)]
    if bn:
        layers.append(nn.BatchNorm2d(out_ch))
    if act:
        layers.append(nn.ReLU(inplace=True))
    return nn.Sequential(*layers)


class BasicBlockV1(nn.Module):
    """v1 cell: conv-bn-relu, conv-bn, (+ projection), relu."""

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        self.body = nn.Sequential(
            conv_bn_act(in_ch, out_ch, 3, stride),
            conv_bn_act(out_ch, out_ch, 3, 1, act=False),
        )
        self.proj = None
        if stride != 1 or in_ch != out_ch:
            self.proj = conv_bn_act(in_ch, out_ch, 1, stride, act=False, bn=False)
        self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        s = x if self.proj is None else self.proj(x)
        return self.act(self.body(x) + s)


class BottleneckV2(nn.Module):
    """v2 cell (reference make_cell_v2, resnet.py:181-231): pre-activation
    3x3(stride) -> 3x3 -> 1x1 with an un-normalised 1x1 projection on the
    first block of a stage; no activation after the add. ``preact=False``
    reproduces the reference's first-block-of-first-stage case (r1 has
    neither BN nor ReLU before its conv)."""

    def __init__(self, in_ch: int, mid_ch: int, out_ch: int,
                 stride: int = 1, preact: bool = True):
        super().__init__()
        self.pre1 = (
            nn.Sequential(nn.BatchNorm2d(in_ch), nn.ReLU(inplace=True))
            if preact
            else nn.Identity()
        )
        self.conv1 = NativeConv2d(in_ch, mid_ch, 3, stride=stride, padding=1)
        self.pre2 = nn.Sequential(nn.BatchNorm2d(mid_ch), nn.ReLU(inplace=True))
        self.conv2 = NativeConv2d(mid_ch, mid_ch, 3, padding=1)
        self.pre3 = nn.Sequential(nn.BatchNorm2d(mid_ch), nn.ReLU(inplace=True))
        self.conv3 = NativeConv2d(mid_ch, out_ch, 1)
        self.proj = None
        if stride != 1 or in_ch != out_ch:
            # reference r4: plain conv on the RAW input (resnet.py:212-219)
            self.proj = NativeConv2d(in_ch, out_ch, 1, stride=stride)

    def forward(self, x):
        y = self.conv1(self.pre1(x))
        y = self.conv2(self.pre2(y))
        y = self.conv3(self.pre3(y))
        s = x if self.proj is None else self.proj(x)
        return s + y


class Head(nn.Module):
    """Global average pool + linear classifier (one cell)."""

    def __init__(self, in_ch: int, num_classes: int, final_bn: bool = False, mknorm=nn.BatchNorm2d):
        super().__init__()
        self.final = (
            nn.Sequential(mknorm(in_ch), nn.ReLU(inplace=True))
            if final_bn
            else nn.Identity()
        )
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(in_ch, num_classes)

    def forward(self, x):
        x = self.pool(self.final(x))
        return self.fc(torch.flatten(x, 1))


def _stem(in_ch: int, filters: int, image_size: int,
          ref_stem: bool = False) -> nn.Module:
    """Stem sized to the image: big images get a stride-2 7x7 + maxpool so
    activations stay tractable (the reference keeps a 3x3 stride-1 stem for
    CIFAR-scale and relies on SP for big images; we keep stride-1 below 128).
    ``ref_stem=True`` forces the reference's stride-1 3x3 stem at ANY size —
    use it when comparing against the reference's published ResNet numbers."""
    if image_size >= 128 and not ref_stem:
        return nn.Sequential(
            NativeConv2d(in_ch, filters, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(filters),
            nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1),
        )
    return conv_bn_act(in_ch, filters, 3, 1)


def get_resnet_v1(
    input_shape,
    num_classes: int = 10,
    n: int = 3,
    num_filters: int = 16,
    device: Optional[str] = None,
    ref_stem: bool = False,
) -> nn.Sequential:
    """6n+2-layer v1 ResNet as a flat Sequential of cells
    (reference resnet.py:145-178)."""
    _, in_ch, H, W = input_shape
    cells = [_stem(in_ch, num_filters, min(H, W), ref_stem)]
    ch = num_filters
    for group in range(3):
        out_ch = num_filters * (2**group)
        for block in range(n):
            stride = 2 if (group > 0 and block == 0) else 1
            cells.append(BasicBlockV1(ch, out_ch, stride))
            ch = out_ch
    cells.append(Head(ch, num_classes))
    model = nn.Sequential(*cells)
    if device is not None:
        model = model.to(device)
    return model


def get_resnet_v2(
    input_shape,
    num_classes: int = 10,
    n: int = 12,
    num_filters: int = 16,
    device: Optional[str] = None,
    ref_stem: bool = False,
) -> nn.Sequential:
    """9n+2-layer v2 (pre-activation bottleneck) ResNet
    (reference resnet.py:270-326)."""
    _, in_ch, H, W = input_shape
    cells = [_stem(in_ch, num_filters, min(H, W), ref_stem)]
    ch = num_filters
    mid = num_filters
    for group in range(3):
        # reference width schedule (resnet.py:288-300): stage 0 expands
        # 4x (mids 16->outs 64), later stages 2x (64->128, 128->256);
        # stage 0 never downsamples
        out = mid * (4 if group == 0 else 2)
        for block in range(n):
            stride = 2 if (group > 0 and block == 0) else 1
            cells.append(BottleneckV2(
                ch, mid, out, stride,
                preact=not (group == 0 and block == 0),
            ))
            ch = out
        mid = out
    cells.append(Head(ch, num_classes, final_bn=True))
    model = nn.Sequential(*cells)
    if device is not None:
        model = model.to(device)
    return model


def get_resnet18_cells(
    input_shape,
    num_classes: int = 1000,
    width: int = 64,
    device: Optional[str] = None,
    ref_stem: bool = False,
) -> nn.Sequential:
    """ImageNet-style ResNet-18 as flat cells (BASELINE config 1:
    ResNet-18 layer parallelism at 224²). Basic-block counts 2-2-2-2."""
    _, in_ch, H, W = input_shape
    cells = [_stem(in_ch, width, min(H, W), ref_stem)]
    ch = width
    for group, blocks in enumerate([2, 2, 2, 2]):
        out = width * (2**group)
        for b in range(blocks):
            stride = 2 if (group > 0 and b == 0) else 1
            cells.append(BasicBlockV1(ch, out, stride))
            ch = out
    cells.append(Head(ch, num_classes, final_bn=False))
    model = nn.Sequential(*cells)
    if device is not None:
        model = model.to(device)
    return model


def get_resnet101_cells(
    input_shape,
    num_classes: int = 1000,
    width: int = 64,
    device: Optional[str] = None,
    ref_stem: bool = False,
) -> nn.Sequential:
    """ImageNet-style ResNet-101 as flat cells (for BASELINE config 4:
    ResNet-101 SP+PP at 2048²). Bottleneck counts 3-4-23-3."""
    _, in_ch, H, W = input_shape
    cells = [_stem(in_ch, width, min(H, W), ref_stem)]
    ch = width
    for group, blocks in enumerate([3, 4, 23, 3]):
        mid = width * (2**group)
        for b in range(blocks):
            stride = 2 if (group > 0 and b == 0) else 1
            cells.append(BottleneckV2(ch, mid, mid * 4, stride))
            ch = mid * 4
    cells.append(Head(ch, num_classes, final_bn=True))
    model = nn.Sequential(*cells)
    if device is not None:
        model = model.to(device)
    return model
