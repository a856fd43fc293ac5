"""act-ckpt drilldown #2: eager-path checkpoint divergence on GPU.

All native ops disabled; pure torch/MIOpen. Finds (a) whether the eager
forward itself is deterministic, (b) the minimal cell prefix whose grads
diverge between plain and checkpointed execution, (c) whether recompute
sees/produces bitwise-identical values via a call-recording wrapper.
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ["MPI4DL_NATIVE_CONV"] = "0"
os.environ["MPI4DL_NATIVE_BN"] = "0"
os.environ["MPI4DL_NATIVE_POOL"] = "0"

import torch
from torch.utils.checkpoint import checkpoint

from mpi4dl_amd.models.amoebanet import amoebanetd

DEV = torch.device("cuda", 0)


def build():
    torch.manual_seed(0)
    m = amoebanetd(100, 6, 64).to(DEV)
    m.train()
    return m


def data():
    torch.manual_seed(42)
    return torch.randn(2, 3, 256, 256, device=DEV)


def tup(x):
    return x if isinstance(x, tuple) else (x,)


def fwd_prefix(model, x, n, ckpt):
    cells = list(model)[:n]
    with torch.autocast("cuda", dtype=torch.bfloat16):
        for c in cells:
            x = checkpoint(c, x, use_reentrant=False) if ckpt else c(x)
    return x


def loss_of(y):
    return sum(t.float().square().mean() for t in tup(y))


def grads(model, n, ckpt):
    m = build()
    x = data()
    y = fwd_prefix(m, x, n, ckpt)
    loss_of(y).backward()
    return float(loss_of(y).detach()), [
        None if p.grad is None else p.grad.detach().float().clone()
        for p in m.parameters()
    ]


def main():
    torch.cuda.init()
    # (a) forward determinism of the full eager model
    m = build()
    x = data()
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        y1 = m(x)
        y2 = m(x)
    print("eager fwd deterministic:", bool((y1 == y2).all().item()))

    # (b) minimal failing prefix
    ncells = len(list(build()))
    first_fail = None
    for n in range(1, ncells + 1):
        la, ga = grads(None, n, False)
        lb, gb = grads(None, n, True)
        bad = sum(
            1
            for a, b in zip(ga, gb)
            if a is not None and not torch.allclose(a, b, rtol=1e-2, atol=1e-3)
        )
        used = sum(1 for a in ga if a is not None)
        print(f"prefix {n}: loss {la:.6f}/{lb:.6f} bad={bad}/{used}", flush=True)
        if bad and first_fail is None:
            first_fail = n
            break

    if first_fail is None:
        print("no failing prefix found?!")
        return

    # (c) record both invocations of each checkpointed cell
    n = first_fail
    m = build()
    x = data()
    cells = list(m)[:n]
    records = [[] for _ in cells]

    def wrap(i, cell):
        def fn(inp):
            out = cell(inp)
            records[i].append(
                (
                    [t.detach().clone() for t in tup(inp)],
                    [t.detach().clone() for t in tup(out)],
                )
            )
            return out

        return fn

    with torch.autocast("cuda", dtype=torch.bfloat16):
        for i, c in enumerate(cells):
            x = checkpoint(wrap(i, c), x, use_reentrant=False)
    loss_of(x).backward()
    for i, rec in enumerate(records):
        print(f"cell {i}: {len(rec)} invocations")
        if len(rec) >= 2:
            (in1, out1), (in2, out2) = rec[0], rec[1]
            for j, (a, b) in enumerate(zip(in1, in2)):
                same = bool((a == b).all().item()) and a.dtype == b.dtype
                print(
                    f"  in[{j}] same={same} dtypes={a.dtype}/{b.dtype} "
                    f"maxdiff={float((a.float()-b.float()).abs().max()):.6f}"
                )
            for j, (a, b) in enumerate(zip(out1, out2)):
                same = bool((a == b).all().item()) and a.dtype == b.dtype
                print(
                    f"  out[{j}] same={same} dtypes={a.dtype}/{b.dtype} "
                    f"maxdiff={float((a.float()-b.float()).abs().max()):.6f}"
                )


if __name__ == "__main__":
    main()
