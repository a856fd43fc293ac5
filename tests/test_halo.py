"""Halo exchange validation (gloo, CPU).

Mirrors the reference's strongest correctness oracle — exact integer
match of exchanged halos against a ground truth built from an arange
image (benchmark_sp_halo_exchange.py:417-578) — and goes further:
forward AND backward parity of HaloConv2d / HaloPool2d against the
undistributed op (the reference only validates forward).
"""

import torch
import torch.nn.functional as F

from dist_util import run_distributed


def _halo_fwd_body(rank, world, slice_method, h):
    from mpi4dl_amd.ops.halo import HaloExchanger, TileLayout, halo_pad

    layout = TileLayout(world, slice_method)
    H = W = 16
    full = (
        torch.arange(2 * 3 * H * W, dtype=torch.float32).reshape(2, 3, H, W) + 1.0
    )
    tile = layout.slice_input(full, rank).contiguous()
    ex = HaloExchanger(layout, rank, lambda t: t)
    out = halo_pad(tile, h, ex)

    # ground truth: pad the FULL image, slice my tile + ring
    fullp = F.pad(full, (h, h, h, h))
    r, c = layout.pos(rank)
    th, tw = H // layout.rows, W // layout.cols
    expect = fullp[:, :, r * th : r * th + th + 2 * h, c * tw : c * tw + tw + 2 * h]
    assert torch.equal(out, expect), f"rank {rank} halo mismatch"
    return True


def test_halo_forward_square():
    run_distributed(_halo_fwd_body, 4, ("square", 2))


def test_halo_forward_vertical():
    run_distributed(_halo_fwd_body, 2, ("vertical", 3))


def test_halo_forward_horizontal():
    run_distributed(_halo_fwd_body, 2, ("horizontal", 1))


def _conv_parity_body(rank, world, slice_method, kernel, stride):
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.spatial_conv import HaloConv2d

    torch.manual_seed(7)
    H = W = 16
    Cin, Cout = 3, 5
    full = torch.randn(2, Cin, H, W)
    layout = TileLayout(world, slice_method)

    conv = HaloConv2d(
        Cin,
        Cout,
        kernel,
        stride=stride,
        num_spatial_parts=world,
        slice_method=slice_method,
        spatial_local_rank=rank,
    )
    # reference single-process conv with identical weights
    ref = torch.nn.Conv2d(Cin, Cout, kernel, stride=stride, padding=(kernel - 1) // 2)
    with torch.no_grad():
        ref.weight.copy_(conv.conv.weight)
        ref.bias.copy_(conv.conv.bias)

    tile = layout.slice_input(full, rank).clone().requires_grad_(True)
    out = conv(tile)

    full_in = full.clone().requires_grad_(True)
    ref_out = ref(full_in)
    expect = layout.slice_input(ref_out, rank)
    assert torch.allclose(out, expect, atol=1e-5), (
        f"rank {rank} fwd mismatch {slice_method} k={kernel} s={stride}"
    )

    # backward parity: upstream grad = arange for determinism
    g_full = torch.arange(ref_out.numel(), dtype=torch.float32).reshape(
        ref_out.shape
    ) / ref_out.numel()
    ref_out.backward(g_full)
    out.backward(layout.slice_input(g_full, rank).contiguous())

    # input grads: exact-mode halo backward must equal sliced full grad
    expect_gin = layout.slice_input(full_in.grad, rank)
    assert torch.allclose(tile.grad, expect_gin, atol=1e-5), (
        f"rank {rank} input-grad mismatch"
    )
    # weight grads: sum over tiles == full weight grad
    import torch.distributed as dist

    wg = conv.conv.weight.grad.clone()
    dist.all_reduce(wg)
    assert torch.allclose(wg, ref.weight.grad, atol=1e-4), f"rank {rank} wgrad"
    return True


def test_conv_parity_square_k3():
    run_distributed(_conv_parity_body, 4, ("square", 3, 1))


def test_conv_parity_square_k5_stride2():
    run_distributed(_conv_parity_body, 4, ("square", 5, 2))


def test_conv_parity_vertical_k3():
    run_distributed(_conv_parity_body, 4, ("vertical", 3, 1))


def test_conv_parity_horizontal_k3_stride2():
    run_distributed(_conv_parity_body, 2, ("horizontal", 3, 2))


def _pool_parity_body(rank, world, kind, kernel, stride, padding):
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.spatial_conv import HaloPool2d

    torch.manual_seed(3)
    H = W = 16
    # all-negative values exercise the -inf vs zero pad boundary semantics
    full = -torch.rand(2, 3, H, W) - 0.5
    layout = TileLayout(world, "square")
    pool = HaloPool2d(
        kind,
        kernel,
        stride=stride,
        padding=padding,
        num_spatial_parts=world,
        slice_method="square",
        spatial_local_rank=rank,
    )
    tile = layout.slice_input(full, rank).contiguous()
    out = pool(tile)
    if kind == "max":
        ref = F.max_pool2d(full, kernel, stride, padding=padding)
    else:
        ref = F.avg_pool2d(full, kernel, stride, padding=padding, count_include_pad=True)
    expect = layout.slice_input(ref, rank)
    assert torch.allclose(out, expect, atol=1e-6), f"rank {rank} {kind} pool mismatch"
    return True


def test_maxpool_parity():
    run_distributed(_pool_parity_body, 4, ("max", 3, 2, 1))


def test_avgpool_parity():
    run_distributed(_pool_parity_body, 4, ("avg", 3, 2, 1))


def test_meta_halo_shapes():
    """Partitioner shape-inference path: meta tensors, no comm."""
    from mpi4dl_amd.ops.spatial_conv import HaloConv2d

    conv = HaloConv2d(3, 8, 3, num_spatial_parts=4, spatial_local_rank=0)
    x = torch.zeros(2, 3, 16, 16, device="meta")
    y = conv.to("meta")(x)
    assert y.shape == (2, 8, 16, 16)


def _overlap_body(rank, world, slice_method):
    """drop-mode overlap path must equal the blocking drop path exactly
    (forward AND weight grads)."""
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.spatial_conv import HaloConv2d

    torch.manual_seed(7)
    H = W = 16
    full = torch.randn(2, 3, H, W)
    layout = TileLayout(world, slice_method)

    def mk(grad_mode):
        torch.manual_seed(1)
        return HaloConv2d(
            3, 4, 3, num_spatial_parts=world, slice_method=slice_method,
            spatial_local_rank=rank, grad_mode=grad_mode,
        )

    conv_overlap = mk("drop")            # takes the overlap path
    conv_block = mk("drop")
    conv_block.grad_mode = "drop"
    # force blocking by pretending exact dispatch conditions fail
    conv_block._forward_overlap = None
    tile = layout.slice_input(full, rank).contiguous()

    t1 = tile.clone().requires_grad_(True)
    y1 = conv_overlap(t1)
    t2 = tile.clone().requires_grad_(True)
    from mpi4dl_amd.ops.halo import halo_pad

    xp = halo_pad(t2, 1, conv_block.exchanger, "drop")
    y2 = conv_block.conv(xp)
    assert torch.allclose(y1, y2, atol=1e-6), (y1 - y2).abs().max()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(t1.grad, t2.grad, atol=1e-5)
    assert torch.allclose(
        conv_overlap.conv.weight.grad, conv_block.conv.weight.grad, atol=1e-5
    )
    return True


def test_halo_overlap_square():
    run_distributed(_overlap_body, 4, ("square",))


def test_halo_overlap_vertical():
    run_distributed(_overlap_body, 2, ("vertical",))


def _exchange_layer_body(rank, world, h, d2):
    """Standalone HaloExchangeLayer (C9): output = tile + ring, exact."""
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.spatial_conv import HaloExchangeLayer

    layout = TileLayout(world, "vertical")
    H = W = 16
    full = torch.arange(1.0 * 2 * 3 * H * W).reshape(2, 3, H, W)
    tile = layout.slice_input(full, rank).contiguous()
    layer = HaloExchangeLayer(
        h, num_spatial_parts=world, slice_method="vertical",
        spatial_local_rank=rank,
    )
    out = layer(tile)
    r, c = layout.pos(rank)
    tw = W // layout.cols
    fullp = F.pad(full, (h, h, h, h))
    expect = fullp[:, :, 0 : H + 2 * h, c * tw : c * tw + tw + 2 * h]
    assert torch.equal(out, expect), f"rank {rank}"
    return True


def test_halo_exchange_layer():
    run_distributed(_exchange_layer_body, 2, (3, False))


def _pool_overlap_body(rank, world, no_overlap):
    import os

    os.environ["MPI4DL_NO_OVERLAP"] = "1" if no_overlap else "0"
    import torch

    from mpi4dl_amd.ops.spatial_conv import HaloPool2d

    torch.manual_seed(0)
    x = torch.randn(2, 4, 16, 16 // world * world)
    from mpi4dl_amd.ops.halo import TileLayout

    layout = TileLayout(world, "vertical")
    tile = layout.slice_input(x, rank).contiguous().requires_grad_(True)
    outs = {}
    for kind, cip in (("max", True), ("avg", False), ("avg", True)):
        pool = HaloPool2d(
            kind, 3, stride=1, padding=1, num_spatial_parts=world,
            slice_method="vertical", spatial_local_rank=rank,
            count_include_pad=cip,
        )
        y = pool(tile)
        g = torch.ones_like(y)
        (gx,) = torch.autograd.grad(y, tile, g)
        outs[(kind, cip)] = (y.detach().tolist(), gx.tolist())
    return outs


def test_pool_overlap_equals_blocking():
    """Stride-1 pool halo overlap == blocking path, forward AND exact
    backward, for max and both avg semantics."""
    a = run_distributed(_pool_overlap_body, 2, (False,))
    b = run_distributed(_pool_overlap_body, 2, (True,))
    for r in range(2):
        for key in a[r]:
            ya, ga = a[r][key]
            yb, gb = b[r][key]
            assert torch.allclose(torch.tensor(ya), torch.tensor(yb),
                                  atol=1e-6), key
            assert torch.allclose(torch.tensor(ga), torch.tensor(gb),
                                  atol=1e-6), key
