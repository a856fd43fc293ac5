"""TileBatchNorm2d — BatchNorm with statistics reduced across the tiles
of one spatial partition.

The reference uses plain ``nn.BatchNorm2d`` inside spatial partitions,
so each tile normalises with ITS OWN statistics — distributed training
silently computes a different function than the undistributed model
(and than the paper's math). This module all-reduces the per-channel
sums over the spatial tile group, making spatially-parallel BN exactly
equal to single-GPU BN (our SP parity tests rely on it).

Works on gloo (CPU tests) and RCCL; autograd-correct via a sum-allreduce
whose adjoint is another sum-allreduce. Falls back to local statistics
when no group is set (plain model) and on meta tensors (shape
inference). torch.nn.SyncBatchNorm is CUDA/NCCL-only and syncs over a
whole group of its own; this version is backend-agnostic and scoped to
the tile group.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist
import torch.nn as nn


class _AllReduceSum(torch.autograd.Function):
    """Differentiable sum-allreduce: forward sums over the group, backward
    sums the incoming grads over the group (the adjoint of broadcast-sum)."""

    @staticmethod
    def forward(ctx, t: torch.Tensor, group):
        ctx.group = group
        out = t.clone()
        dist.all_reduce(out, group=group)
        return out

    @staticmethod
    def backward(ctx, g: torch.Tensor):
        g = g.contiguous().clone()
        dist.all_reduce(g, group=ctx.group)
        return g, None


def allreduce_sum(t: torch.Tensor, group) -> torch.Tensor:
    return _AllReduceSum.apply(t, group)


class TileBatchNorm2d(nn.BatchNorm2d):
    """Drop-in BatchNorm2d whose batch statistics span the tile group."""

    def __init__(self, num_features: int, group=None, relu: bool = False, **kw):
        super().__init__(num_features, **kw)
        self.group = group
        self.relu = relu  # fused BN+ReLU (native GPU path only)

    def _use_sync(self, x) -> bool:
        return (
            self.training
            and self.group is not None
            and not x.is_meta
            and dist.is_initialized()
        )

    def forward(self, x):
        if (
            x.is_cuda
            and not x.is_meta
            and (x.shape[-2] * x.shape[-1]) % 8 == 0
            and os.environ.get("MPI4DL_NATIVE_BN", "1") != "0"
        ):
            return self._forward_native(x)
        if not self._use_sync(x):
            y = super().forward(x)
            return torch.relu_(y) if self.relu else y
        n_local = x.numel() // x.shape[1]
        world = dist.get_world_size(group=self.group)
        n = n_local * world  # tiles are equal-sized (power-of-two constraint)
        # two-pass variance: matches torch BN's centred formula closely
        # (the one-pass E[x^2]-mean^2 form loses ~1e-6 relative precision,
        # which measurably perturbs long BN chains)
        mean = allreduce_sum(x.sum(dim=(0, 2, 3)), self.group) / n
        centred = x - mean.view(1, -1, 1, 1)
        var = allreduce_sum((centred * centred).sum(dim=(0, 2, 3)), self.group) / n
        if self.track_running_stats:
            with torch.no_grad():
                m = self.momentum if self.momentum is not None else 0.1
                self.running_mean.mul_(1 - m).add_(mean.detach(), alpha=m)
                unbiased = var.detach() * (n / max(n - 1, 1))
                self.running_var.mul_(1 - m).add_(unbiased, alpha=m)
                self.num_batches_tracked += 1
        inv = torch.rsqrt(var + self.eps)
        out = centred * inv.view(1, -1, 1, 1)
        if self.affine:
            out = out * self.weight.view(1, -1, 1, 1) + self.bias.view(1, -1, 1, 1)
        return torch.relu(out) if self.relu else out

    # -- gemscore fused path (MI355X) ----------------------------------

    def _forward_native(self, x):
        from . import backend
        from .native import native_batchnorm

        ge = backend.ext()
        x = x.contiguous()
        C = self.num_features
        n = x.numel() // C
        group = self.group if (self.group is not None and dist.is_initialized()) else None
        if self.training:
            # persistent zeroed fp64 buffer: stats64 accumulates into it
            # and bn_finalize(rezero=True) restores the zero invariant —
            # removes the per-call torch.zeros fill (~3k tiny launches
            # per bench step)
            buf = getattr(self, "_stats64", None)
            if buf is None or buf.device != x.device or buf.numel() != 2 * C:
                buf = torch.zeros(2 * C, device=x.device, dtype=torch.float64)
                self._stats64 = buf
            stats = ge.bn_stats64_acc(x, buf)
            if group is not None:
                dist.all_reduce(stats, group=group)
                n *= dist.get_world_size(group=group)
            m = self.momentum if self.momentum is not None else 0.1
            mv = ge.bn_finalize(
                stats,
                self.running_mean if self.track_running_stats else None,
                self.running_var if self.track_running_stats else None,
                self.num_batches_tracked if self.track_running_stats else None,
                m, float(n), self.eps, rezero=True,
            )
            mean, invstd = mv[:C], mv[C:]
        else:
            mean = self.running_mean.float().contiguous()
            invstd = torch.rsqrt(self.running_var.float() + self.eps).contiguous()
        return native_batchnorm(
            x, self.weight, self.bias, mean.contiguous(), invstd.contiguous(),
            group, float(n), self.training, self.relu,
        )
