"""Autograd wrappers over the gemscore CDNA4 kernels.

Each Function pairs a hand-written HIP forward with its hand-written
backward (no eager fallback on GPU — backend.py raises if the extension
is missing). CPU paths never reach these; the modules in
ops/spatial_conv.py and ops/norm.py dispatch here only for CUDA inputs.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from . import backend


class MaxPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p):
        ge = backend.ext()
        x = x.contiguous()
        y, idx = ge.maxpool_fwd(x, k, s, p)
        ctx.save_for_backward(idx)
        ctx.geom = (x.shape[-2], x.shape[-1], k, s, p)
        return y

    @staticmethod
    def backward(ctx, go):
        (idx,) = ctx.saved_tensors
        H, W, k, s, p = ctx.geom
        gi = backend.ext().maxpool_bwd(go.contiguous(), idx, H, W, k, s, p)
        return gi, None, None, None


def native_maxpool(x, k, s, p):
    return MaxPool2dFn.apply(x, k, s, p)


class AvgPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p, gr0, gc0, Hg, Wg, include_pad):
        ge = backend.ext()
        x = x.contiguous()
        y = ge.avgpool_fwd(x, k, s, p, gr0, gc0, Hg, Wg, include_pad)
        ctx.geom = (x.shape[-2], x.shape[-1], k, s, p, gr0, gc0, Hg, Wg, include_pad)
        return y

    @staticmethod
    def backward(ctx, go):
        H, W, k, s, p, gr0, gc0, Hg, Wg, include_pad = ctx.geom
        gi = backend.ext().avgpool_bwd(
            go.contiguous(), H, W, k, s, p, gr0, gc0, Hg, Wg, include_pad
        )
        return (gi,) + (None,) * 8


def native_avgpool(x, k, s, p, gr0=None, gc0=None, Hg=None, Wg=None, include_pad=True):
    if gr0 is None:
        gr0, gc0 = -p, -p
        Hg, Wg = x.shape[-2], x.shape[-1]
    return AvgPool2dFn.apply(x, k, s, p, gr0, gc0, Hg, Wg, include_pad)


class BatchNormFn(torch.autograd.Function):
    """Fused (sync-capable) BatchNorm with optional ReLU.

    mean/invstd are precomputed constants (from bn_stats64 + optional
    group allreduce); backward implements the full sync-BN gradient:
    the gsum/gxsum reduction terms are allreduced over the tile group,
    weight/bias grads stay LOCAL (the engine's spatial-group SUM
    allreduce aggregates them with the other parameter grads).
    """

    @staticmethod
    def forward(ctx, x, weight, bias, mean, invstd, group, n_global, training, relu):
        ge = backend.ext()
        x = x.contiguous()
        w32 = weight.detach().float().contiguous()
        b32 = bias.detach().float().contiguous()
        y = ge.bn_apply(x, mean, invstd, w32, b32, relu)
        ctx.save_for_backward(x, y, mean, invstd, w32)
        ctx.wdtype = weight.dtype
        ctx.group = group
        ctx.n_global = n_global
        ctx.training = training
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, go):
        ge = backend.ext()
        x, y, mean, invstd, w32 = ctx.saved_tensors
        go = go.contiguous()
        C = x.shape[1]
        stats = ge.bn_bwd_stats(go, x, y, mean, invstd, ctx.relu)  # fp64
        sf = stats.float()  # ONE cast; gw/gb are views of it
        gw = sf[C:]  # local sum(go*xhat)
        gb = sf[:C]  # local sum(go)
        if ctx.training:
            if ctx.group is not None:
                g_global = stats.clone()
                dist.all_reduce(g_global, group=ctx.group)
                gf = g_global.float()
            else:
                gf = sf
            gi = ge.bn_bwd_apply(
                go, x, y, mean, invstd, w32,
                gf[:C].contiguous(), gf[C:].contiguous(),
                float(ctx.n_global), ctx.relu,
            )
        else:
            # eval: mean/var are constants -> gi = go_eff * w * invstd
            zeros = torch.zeros(2 * C, device=x.device, dtype=torch.float32)
            gi = ge.bn_bwd_apply(
                go, x, y, mean, invstd, w32,
                zeros[:C].contiguous(), zeros[C:].contiguous(),
                1.0, ctx.relu,
            )
        return (gi, gw.to(ctx.wdtype), gb.to(ctx.wdtype)) + (None,) * 6


def native_batchnorm(x, weight, bias, mean, invstd, group, n_global, training, relu=False):
    return BatchNormFn.apply(
        x, weight, bias, mean, invstd, group, n_global, training, relu
    )
