"""Fused elementwise glue for the AmoebaNet cell epilogue.

Reference behaviour replaced: the cell's per-pair sums and final
concat (/root/reference/src/models/amoebanet.py:449-533, Cell.forward's
``torch.cat(states[...])``). The reference (and round-1) cell does
``states.append(h1 + h2)`` per genotype pair and then
``torch.cat([states[i] for i in concat], 1)`` —
at 2048^2 that cat alone re-reads and re-writes the whole cell output
(~9% of the step was such eager glue in profiles/r01_*). AddCat writes
each concat slice ONCE: sum slices compute ``h1 + h2`` directly into
their channel range of the output buffer, passthrough slices copy.
Backward is free: every input's gradient is a channel-narrow VIEW of
the incoming gradient (no kernels).

Works on CPU and GPU (torch.add into a strided out); autograd-correct
including states that feed both the concat and later ops (grad
contributions accumulate via the normal autograd sum).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

_SENTINEL = None


class _AddCatFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, spec, *tensors):
        # spec: list of (a_idx, b_idx_or_-1, channels)
        first = tensors[0]
        n, _, h, w = first.shape
        ctot = sum(c for _, _, c in spec)
        dtype = tensors[0].dtype
        for t in tensors:
            dtype = torch.promote_types(dtype, t.dtype)
        out = torch.empty(n, ctot, h, w, device=first.device, dtype=dtype)
        off = 0
        for a_i, b_i, c in spec:
            sl = out.narrow(1, off, c)
            a = tensors[a_i]
            if b_i < 0:
                sl.copy_(a)
            else:
                torch.add(a, tensors[b_i], out=sl)
            off += c
        ctx.spec = spec
        ctx.n_inputs = len(tensors)
        return out

    @staticmethod
    def backward(ctx, go):
        grads: List[Optional[torch.Tensor]] = [None] * ctx.n_inputs
        off = 0
        for a_i, b_i, c in ctx.spec:
            g = go.narrow(1, off, c)
            grads[a_i] = g if grads[a_i] is None else grads[a_i] + g
            if b_i >= 0:
                grads[b_i] = g if grads[b_i] is None else grads[b_i] + g
            off += c
        return (None, *grads)


def add_cat(entries: List[Tuple[torch.Tensor, Optional[torch.Tensor]]]):
    """entries: per concat slice, (a, b) -> slice = a + b, or (a, None)
    -> slice = a. Returns the channel-concatenated tensor."""
    tensors: List[torch.Tensor] = []
    index = {}

    def idx(t):
        k = id(t)
        if k not in index:
            index[k] = len(tensors)
            tensors.append(t)
        return index[k]

    spec = tuple(
        (idx(a), idx(b) if b is not None else -1, a.shape[1]) for a, b in entries
    )
    return _AddCatFn.apply(spec, *tensors)
