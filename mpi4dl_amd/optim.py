"""FusedSGD: whole-model momentum SGD in one CDNA4 kernel.

The profile of the eager path showed ~5.8k tiny per-parameter kernels
per step (torch SGD iterates parameters). Here the model's parameters
and gradients live as views into single flat fp32 buffers (grads via
comm.FlatGrads — shared with GradReducer so collectives stay single
messages), plus one flat momentum buffer; step() is ONE
gemscore.sgd_momentum launch that also zeroes the grads.

CPU fallback implements the same math with three tensor ops.
"""

from __future__ import annotations

import torch

from .comm import FlatGrads


def flatten_params(module: torch.nn.Module) -> torch.Tensor:
    """Re-home all parameters as views into one flat fp32 buffer
    (pad to a multiple of 4 for float4 kernels)."""
    params = [p for p in module.parameters() if p.requires_grad]
    if not params:
        return torch.zeros(4)
    dev = params[0].device
    total = sum(p.numel() for p in params)
    total_pad = (total + 3) // 4 * 4
    flat = torch.zeros(total_pad, device=dev, dtype=torch.float32)
    off = 0
    for p in params:
        n = p.numel()
        flat[off : off + n].copy_(p.data.reshape(-1))
        p.data = flat[off : off + n].view_as(p)
        off += n
    return flat


class FusedSGD:
    """drop-in minimal optimizer (step / zero_grad) over flat buffers."""

    def __init__(
        self,
        module: torch.nn.Module,
        lr: float = 0.01,
        momentum: float = 0.9,
        weight_decay: float = 0.0,
    ):
        self.module = module
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.flat_p = flatten_params(module)
        self.fg = FlatGrads.get(module, pad_to=self.flat_p.numel())
        self.flat_m = torch.zeros_like(self.flat_p)
        # mirror torch API bits engines touch
        self.param_groups = [
            {"lr": lr, "momentum": momentum, "weight_decay": weight_decay}
        ]

    @property
    def flat_g(self):
        return self.fg.buffer

    def step(self):
        lr = self.param_groups[0]["lr"]
        p, g, m = self.flat_p, self.flat_g, self.flat_m
        if p.is_cuda:
            from .ops import backend

            backend.ext().sgd_momentum(
                p, g, m, lr, self.momentum, self.weight_decay
            )
            return
        if self.weight_decay:
            g = g.add(p, alpha=self.weight_decay)
        m.mul_(self.momentum).add_(g)
        p.add_(m, alpha=-lr)
        self.fg.zero_()

    def zero_grad(self, set_to_none: bool = False):
        # the GPU kernel already zeroed grads in step(); CPU path zeroed
        # in step() as well — keep idempotent for engine.update()
        if not self.flat_p.is_cuda:
            self.fg.zero_()

    def state_dict(self):
        return {
            "momentum_buffer": self.flat_m,
            "lr": self.lr,
            "momentum": self.momentum,
            "weight_decay": self.weight_decay,
        }

    def load_state_dict(self, sd):
        self.flat_m.copy_(sd["momentum_buffer"])
