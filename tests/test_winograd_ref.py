"""Winograd F(2x2,3x3) oracle vs direct convolution — the numerical
reference the round-2 MFMA Winograd kernel will be tested against."""

import pytest
import torch
import torch.nn.functional as F

from dist_util import run_distributed

from mpi4dl_amd.ops.winograd_ref import filter_transform, winograd_conv2d_ref


@pytest.mark.parametrize(
    "N,C,K,H,W,pad",
    [
        (1, 1, 1, 4, 4, 1),
        (2, 3, 8, 16, 16, 1),
        (1, 4, 4, 15, 17, 1),   # odd sizes: tile-pad + crop
        (1, 8, 16, 9, 9, 0),    # valid conv
        (3, 16, 32, 32, 32, 1),
    ],
)
def test_matches_direct_conv(N, C, K, H, W, pad):
    torch.manual_seed(0)
    x = torch.randn(N, C, H, W)
    w = torch.randn(K, C, 3, 3) * 0.2
    b = torch.randn(K)
    ref = F.conv2d(x, w, b, stride=1, padding=pad)
    got = winograd_conv2d_ref(x, w, b, padding=pad)
    assert got.shape == ref.shape, (got.shape, ref.shape)
    err = (got - ref).abs().max() / ref.abs().max()
    assert err < 1e-5, float(err)


def test_bf16_inputs_fp32_transforms():
    """The kernel's numeric plan: bf16 data, fp32 transforms/accum —
    error must stay within direct bf16 conv tolerance."""
    torch.manual_seed(1)
    x = torch.randn(2, 16, 32, 32)
    w = torch.randn(32, 16, 3, 3) * 0.1
    ref = F.conv2d(x, w, None, stride=1, padding=1)
    got = winograd_conv2d_ref(
        x.to(torch.bfloat16).float(), w.to(torch.bfloat16).float(), None, 1
    )
    rel = (got - ref).abs().max() / ref.abs().max()
    assert rel < 0.03, float(rel)


def test_filter_transform_shape_and_linearity():
    torch.manual_seed(2)
    w1 = torch.randn(4, 3, 3, 3)
    w2 = torch.randn(4, 3, 3, 3)
    U = filter_transform(w1)
    assert U.shape == (4, 3, 4, 4)
    # transform is linear: U(w1 + w2) = U(w1) + U(w2)
    assert torch.allclose(
        filter_transform(w1 + w2), U + filter_transform(w2), atol=1e-6
    )


@pytest.mark.parametrize(
    "N,C,K,H,W,pad", [(2, 8, 16, 17, 15, 1), (1, 16, 8, 32, 32, 1),
                      (2, 4, 4, 10, 10, 0)]
)
def test_bmm_variant_matches_direct(N, C, K, H, W, pad):
    """The batched-GEMM formulation (the one that rides hipBLASLt on
    GPU) must match direct conv too."""
    from mpi4dl_amd.ops.winograd_ref import winograd_bmm_conv2d

    torch.manual_seed(3)
    x = torch.randn(N, C, H, W)
    w = torch.randn(K, C, 3, 3) * 0.2
    b = torch.randn(K)
    ref = F.conv2d(x, w, b, stride=1, padding=pad)
    got = winograd_bmm_conv2d(x, filter_transform(w), b, padding=pad)
    assert got.shape == ref.shape
    err = (got - ref).abs().max() / ref.abs().max()
    assert err < 1e-5, float(err)


def test_winograd_dispatch_env(monkeypatch):
    """MPI4DL_WINOGRAD=1 routes NativeConv2d 3x3/s1 through the
    batched-GEMM winograd path — forward AND backward must match
    nn.Conv2d (backward falls out of autograd over the transforms)."""
    monkeypatch.setenv("MPI4DL_WINOGRAD", "1")
    from mpi4dl_amd.ops.conv_native import NativeConv2d

    torch.manual_seed(0)
    m = NativeConv2d(8, 16, 3, padding=1)
    ref = torch.nn.Conv2d(8, 16, 3, padding=1)
    ref.load_state_dict(m.state_dict())
    x = torch.randn(2, 8, 16, 16, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y, y2 = m(x), ref(x2)
    assert (y - y2).abs().max() < 1e-5
    g = torch.randn_like(y)
    y.backward(g)
    y2.backward(g)
    assert (x.grad - x2.grad).abs().max() < 1e-5
    assert (m.weight.grad - ref.weight.grad).abs().max() < 1e-3
    assert (m.bias.grad - ref.bias.grad).abs().max() < 1e-3
    # 5x5 kernels must NOT take the winograd path
    m5 = NativeConv2d(4, 4, 5, padding=2)
    r5 = torch.nn.Conv2d(4, 4, 5, padding=2)
    r5.load_state_dict(m5.state_dict())
    xx = torch.randn(1, 4, 12, 12)
    assert (m5(xx) - r5(xx)).abs().max() < 1e-6


def _sp_winograd_body(rank, world, steps, batch, parts, lr):
    import os

    os.environ["MPI4DL_WINOGRAD"] = "1"
    import test_spatial_engine as T

    return T._spatial_body(
        rank, world, steps, batch, parts, lr, "vertical", 2, 1, 2, 1
    )


def test_winograd_composes_with_spatial_engine():
    """MPI4DL_WINOGRAD=1 under the full SP engine (halo-padded 3x3
    convs route through the batched-GEMM path): trajectory still tracks
    serial training."""
    import test_spatial_engine as T

    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = T._serial_losses(steps, batch, parts, lr)
    got = run_distributed(_sp_winograd_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 5e-4, (expected, got)
