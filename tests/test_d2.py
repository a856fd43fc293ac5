"""D2 (fused halo) validation: with BN in eval mode the D2 forward must
equal the serial model EXACTLY (the fused exchange only re-routes where
neighbour pixels come from); in train mode the trajectory stays close
(surplus pixels enter BN statistics)."""

import torch

from dist_util import run_distributed

IMG = 32
NCLS = 10
N = 3
F = 8


def _d2_eval_body(rank, world, fused, method="vertical"):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models import resnet, resnet_spatial_d2
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.plan import SpatialPlan

    comm = Communicator(
        split_size=1,
        ENABLE_SPATIAL=True,
        num_spatial_parts=world,
        spatial_size=1,
        backend="gloo",
    )
    torch.manual_seed(0)
    serial = resnet.get_resnet_v2((1, 3, IMG, IMG), NCLS, n=N, num_filters=F)
    ncells = len(serial)
    plan = SpatialPlan(comm, [ncells], method)
    torch.manual_seed(0)
    d2 = resnet_spatial_d2.get_resnet_v2(
        (1, 3, IMG, IMG), NCLS, n=N, num_filters=F, plan=plan, fused_layers=fused
    )
    serial.eval()
    d2.eval()
    torch.manual_seed(9)
    x = torch.randn(2, 3, IMG, IMG)
    layout = TileLayout(world, method)
    with torch.no_grad():
        y_ser = x
        for cell in list(serial)[:-1]:
            y_ser = cell(y_ser)
        y = layout.slice_input(x, rank).contiguous()
        for cell in list(d2)[:-1]:
            y = cell(y)
    expect = layout.slice_input(y_ser, rank)
    assert y.shape == expect.shape, (y.shape, expect.shape)
    assert torch.allclose(y, expect, atol=1e-5), (
        f"rank {rank}: max err {(y - expect).abs().max()}"
    )
    return True


def test_d2_eval_exact_fused2():
    run_distributed(_d2_eval_body, 2, (2,))


def test_d2_eval_exact_fused4():
    run_distributed(_d2_eval_body, 2, (4,))


def test_d2_eval_exact_square():
    """D2 fused halo on a 2x2 square grid: both axes pad interior sides
    only and the corner strips route through the fused exchange."""
    run_distributed(_d2_eval_body, 4, (2, "square"))


def _d2_train_body(rank, world, steps):
    """Full SP engine with the D2 model: loss finite and decreasing-ish."""
    import torch.nn as nn

    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial_d2
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    split = 2
    comm = Communicator(
        split_size=split,
        ENABLE_SPATIAL=True,
        num_spatial_parts=2,
        spatial_size=1,
        backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial_d2.get_resnet_v2((1, 3, IMG, IMG), NCLS, n=N, num_filters=F)
    ncells = len(probe)
    # the boundary must fall where surplus == 0: after the last block of a
    # fused group (the head cell alone forms the LP stage)
    balance = [ncells - 1, 1]
    plan = SpatialPlan(comm, balance, "vertical")
    torch.manual_seed(0)
    model = resnet_spatial_d2.get_resnet_v2(
        (1, 3, IMG, IMG), NCLS, n=N, num_filters=F, plan=plan, fused_layers=2
    )
    gen = model_generator(model, split, (1, 3, IMG, IMG), balance=balance)
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    eng = train_model_spatial(
        gen, comm.local_rank, 2, 2, comm, slice_method="vertical",
        device=torch.device("cpu"),
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    x = torch.randn(2, 3, IMG, IMG)
    y = torch.randint(0, NCLS, (2,))
    losses = []
    for _ in range(steps):
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_d2_training_runs():
    out = run_distributed(_d2_train_body, 3, (4,))
    losses = out[-1]
    assert all(abs(l) < 1e4 for l in losses)
    # same batch every step with SGD: loss should drop
    assert losses[-1] < losses[0], losses


def _amoeba_d2_eval_body(rank, world):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.amoebanet import amoebanetd
    from mpi4dl_amd.models.amoebanet_d2 import amoebanetd_d2
    from mpi4dl_amd.ops.halo import TileLayout
    from mpi4dl_amd.ops.plan import SpatialPlan

    comm = Communicator(
        split_size=1, ENABLE_SPATIAL=True, num_spatial_parts=world,
        spatial_size=1, backend="gloo",
    )
    torch.manual_seed(0)
    serial = amoebanetd(NCLS, 3, 32)
    ncells = len(serial)
    plan = SpatialPlan(comm, [ncells], "vertical")
    torch.manual_seed(0)
    d2 = amoebanetd_d2(NCLS, 3, 32, plan=plan)
    serial.eval(); d2.eval()
    torch.manual_seed(9)
    x = torch.randn(1, 3, 64, 64)
    layout = TileLayout(world, "vertical")
    with torch.no_grad():
        # deeper cells' tiles shrink below the 1x7 halo (3) — validate the
        # cells where tiling is geometrically valid (stem + 2 reductions +
        # first normal cell; real configs keep SP on early cells only)
        ys = x
        for cell in list(serial)[:4]:
            ys = cell(ys)
        yt = layout.slice_input(x, rank).contiguous()
        for cell in list(d2)[:4]:
            yt = cell(yt)
    for a, b in zip(
        yt if isinstance(yt, tuple) else (yt,),
        ys if isinstance(ys, tuple) else (ys,),
    ):
        expect = layout.slice_input(b, rank)
        assert a.shape == expect.shape, (a.shape, expect.shape)
        assert torch.allclose(a, expect, atol=1e-4), (
            f"rank {rank} max err {(a - expect).abs().max()}"
        )
    return True


def test_amoebanet_d2_eval_exact():
    run_distributed(_amoeba_d2_eval_body, 2, ())
