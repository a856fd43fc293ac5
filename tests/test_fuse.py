"""AddCat (ops/fuse.py) — value and gradient parity vs the eager
h1+h2 / torch.cat formulation, including shared inputs (a tensor that
feeds both a sum slice and later ops) and copy slices."""

import torch

from mpi4dl_amd.ops.fuse import add_cat


def _eager(entries):
    return torch.cat(
        [a if b is None else a + b for a, b in entries], dim=1
    )


def test_add_cat_values_and_grads():
    torch.manual_seed(0)
    a = torch.randn(2, 3, 4, 4, requires_grad=True)
    b = torch.randn(2, 3, 4, 4, requires_grad=True)
    c = torch.randn(2, 5, 4, 4, requires_grad=True)
    d = torch.randn(2, 5, 4, 4, requires_grad=True)
    e = torch.randn(2, 2, 4, 4, requires_grad=True)

    entries = [(a, b), (c, d), (e, None)]
    out = add_cat(entries)
    ref = _eager(entries)
    assert torch.equal(out, ref)

    g = torch.randn_like(out)
    out.backward(g)
    grads = [t.grad.clone() for t in (a, b, c, d, e)]
    for t in (a, b, c, d, e):
        t.grad = None
    ref.backward(g)
    for got, t in zip(grads, (a, b, c, d, e)):
        assert torch.equal(got, t.grad)


def test_add_cat_shared_tensor():
    """One tensor used in TWO slices (sum + copy) accumulates both
    gradient contributions."""
    torch.manual_seed(1)
    a = torch.randn(1, 4, 2, 2, requires_grad=True)
    b = torch.randn(1, 4, 2, 2, requires_grad=True)
    entries = [(a, b), (a, None)]
    out = add_cat(entries)
    ref = _eager(entries)
    assert torch.equal(out, ref)
    out.sum().backward()
    ga, gb = a.grad.clone(), b.grad.clone()
    a.grad = b.grad = None
    ref.sum().backward()
    assert torch.equal(ga, a.grad)
    assert torch.equal(gb, b.grad)


def test_add_cat_dtype_promotion():
    a = torch.randn(1, 2, 2, 2, dtype=torch.bfloat16)
    b = torch.randn(1, 2, 2, 2, dtype=torch.float32)
    out = add_cat([(a, b)])
    assert out.dtype == torch.float32
    assert torch.equal(out, a.float() + b)


def test_add_cat_downstream_use():
    """A sum state consumed by a later op AND by the cat (the
    non-cat-only path materialises it; fused entries still match)."""
    torch.manual_seed(2)
    a = torch.randn(1, 3, 4, 4, requires_grad=True)
    b = torch.randn(1, 3, 4, 4, requires_grad=True)
    w = torch.randn(3, 3, 1, 1, requires_grad=True)

    def run(fused):
        s = a + b                  # materialised state
        t = torch.nn.functional.conv2d(s, w)
        if fused:
            return add_cat([(s, None), (t, None)])
        return torch.cat([s, t], dim=1)

    out = run(True)
    ref = run(False)
    assert torch.equal(out, ref)
    out.sum().backward()
    ga = a.grad.clone()
    wg = w.grad.clone()
    a.grad = w.grad = b.grad = None
    ref.sum().backward()
    assert torch.equal(ga, a.grad)
    assert torch.equal(wg, w.grad)
