"""SP(+LP) engine parity on gloo: the distributed spatial training
trajectory must exactly track serial single-process training
(goes beyond the reference, which only
validates halo forwards)."""

import torch
import torch.nn as nn

from dist_util import run_distributed

IMG = 32
NCLS = 10


def _serial_losses(steps, batch, parts, lr):
    from mpi4dl_amd.models.resnet import get_resnet_v1

    torch.manual_seed(0)
    model = get_resnet_v1((batch, 3, IMG, IMG), num_classes=NCLS, n=1, num_filters=8)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        total = 0.0
        for px, py in zip(x.chunk(parts), y.chunk(parts)):
            loss = crit(model(px).float(), py)
            (loss / parts).backward()
            total += float(loss.detach())
        opt.step()
        opt.zero_grad(set_to_none=False)
        losses.append(total / parts)
    return losses


def _spatial_body(
    rank, world, steps, batch, parts, lr, slice_method, nsp, spatial_size, split, ldp
):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial, verify_spatial_config

    comm = Communicator(
        split_size=split,
        ENABLE_SPATIAL=True,
        num_spatial_parts=nsp,
        spatial_size=spatial_size,
        LOCAL_DP_LP=ldp,
        backend="gloo",
    )
    nsp_list = comm.spatial_parts
    verify_spatial_config(slice_method, IMG, nsp_list)

    # balance must match the plain 5-cell model (n=1): stem + 3 blocks + head
    balance = None
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    ncells = len(probe)
    base, rem = divmod(ncells, split)
    balance = [base + (1 if i < rem else 0) for i in range(split)]

    plan = SpatialPlan(comm, balance, slice_method)
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(
        model, split, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))

    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen,
        local_rank=comm.local_rank,
        batch_size=batch,
        parts=parts,
        comm=comm,
        slice_method=slice_method,
        optimizer=opt,
        device=torch.device("cpu"),
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_sp_vertical_2tiles_plus_lp():
    steps, batch, parts, lr = 3, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # mp = split 2 + 2 tiles - 1 = 3 ranks; last rank computes loss
    got = run_distributed(
        _spatial_body, 3, (steps, batch, parts, lr, "vertical", 2, 1, 2, 1)
    )[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def test_sp_square_4tiles_plus_lp():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # mp = split 2 + 4 tiles - 1 = 5 ranks
    got = run_distributed(
        _spatial_body, 5, (steps, batch, parts, lr, "square", 4, 1, 2, 1)
    )[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def test_sp_skewed_4_to_2():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # spatial partitions [4, 2] + 1 LP stage: mp = 3 + 4 + 2 - 2 = 7 ranks
    got = run_distributed(
        _spatial_body, 7, (steps, batch, parts, lr, "vertical", [4, 2], 2, 3, 1)
    )[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def test_sp_local_dp():
    steps, batch, parts, lr = 2, 4, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # 2 tiles + 1 LP partition x 2 local-DP = mp 2 + 2*2 - ... split=2,
    # nsp=2, spatial_size=1, LDP=2 -> mp = 2 + 2 - 1 + (2-1)*(2-1) = 4
    got = run_distributed(
        _spatial_body, 4, (steps, batch, parts, lr, "vertical", 2, 1, 2, 2)
    )
    # last two ranks are the local-DP pair; each reports shard loss —
    # their mean must match the serial loss
    for e, g0, g1 in zip(expected, got[2], got[3]):
        assert abs(e - (g0 + g1) / 2) < 2e-4, (expected, got[2], got[3])


def _spatial_1f1b_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    comm = Communicator(
        split_size=2, ENABLE_SPATIAL=True, num_spatial_parts=2,
        spatial_size=1, backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    ncells = len(probe)
    base, rem = divmod(ncells, 2)
    balance = [base + (1 if i < rem else 0) for i in range(2)]
    plan = SpatialPlan(comm, balance, "vertical")
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(
        model, 2, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm, slice_method="vertical",
        optimizer=opt, device=torch.device("cpu"), schedule="1f1b",
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_sp_1f1b_parity():
    """1F1B (PipeDream-flush) schedule composed with spatial tiles must
    produce the identical trajectory to serial training (same gradients
    as GPipe, different interleaving)."""
    steps, batch, parts, lr = 2, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_spatial_1f1b_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def test_sp_horizontal_2tiles_plus_lp():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(
        _spatial_body, 3, (steps, batch, parts, lr, "horizontal", 2, 1, 2, 1)
    )[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def _sp_eval_body(rank, world, batch):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    comm = Communicator(
        split_size=2, ENABLE_SPATIAL=True, num_spatial_parts=2,
        spatial_size=1, backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    base, rem = divmod(len(probe), 2)
    balance = [base + (1 if i < rem else 0) for i in range(2)]
    plan = SpatialPlan(comm, balance, "vertical")
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(model, 2, input_size=(batch, 3, IMG, IMG),
                          balance=balance)
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    eng = train_model_spatial(
        gen, comm.local_rank, batch, 1, comm, slice_method="vertical",
        device=torch.device("cpu"),
    )
    torch.manual_seed(9)
    x = torch.randn(batch, 3, IMG, IMG)
    y = torch.randint(0, NCLS, (batch,))
    loss, corr, seen = eng.run_eval(x, y)
    return loss, corr, seen


def test_sp_eval_parity():
    """Spatial run_eval (forward-only, eval-mode BN) must match serial
    model.eval() on the full image — the spatial input slicing now runs
    in run_eval too."""
    from mpi4dl_amd.models.resnet import get_resnet_v1

    batch = 2
    torch.manual_seed(0)
    ref = get_resnet_v1((batch, 3, IMG, IMG), num_classes=NCLS, n=1, num_filters=8)
    ref.eval()
    torch.manual_seed(9)
    x = torch.randn(batch, 3, IMG, IMG)
    y = torch.randint(0, NCLS, (batch,))
    with torch.no_grad():
        logits = ref(x).float()
        rl = float(nn.functional.cross_entropy(logits, y))
        rcorr = int((logits.argmax(1) == y).sum())
    got = run_distributed(_sp_eval_body, 3, (batch,))
    loss, corr, seen = got[-1]
    assert seen == batch and corr == rcorr
    assert abs(loss - rl) < 1e-4, (loss, rl)


def _sp_act_ckpt_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    comm = Communicator(
        split_size=2, ENABLE_SPATIAL=True, num_spatial_parts=2,
        spatial_size=1, backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    base, rem = divmod(len(probe), 2)
    balance = [base + (1 if i < rem else 0) for i in range(2)]
    plan = SpatialPlan(comm, balance, "vertical")
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(
        model, 2, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm, slice_method="vertical",
        optimizer=opt, device=torch.device("cpu"), act_ckpt=True,
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_sp_act_ckpt_parity():
    """Activation checkpointing with SPATIAL cells: halo exchanges re-run
    during recompute and must pair up across tile ranks (symmetric
    backward schedule). Trajectory must still match serial exactly."""
    steps, batch, parts, lr = 2, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_sp_act_ckpt_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def _sp_drop_body(rank, world, steps, batch, parts, lr, no_overlap):
    import os

    os.environ["MPI4DL_NO_OVERLAP"] = "1" if no_overlap else "0"
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    comm = Communicator(
        split_size=2, ENABLE_SPATIAL=True, num_spatial_parts=2,
        spatial_size=1, backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    base, rem = divmod(len(probe), 2)
    balance = [base + (1 if i < rem else 0) for i in range(2)]
    plan = SpatialPlan(comm, balance, "vertical", grad_mode="drop")
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(
        model, 2, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm, slice_method="vertical",
        optimizer=opt, grad_mode="drop", device=torch.device("cpu"),
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_drop_overlap_equals_blocking():
    """grad_mode='drop': the halo/compute-overlap trajectory must be
    identical to the blocking-exchange trajectory (MPI4DL_NO_OVERLAP=1)
    — overlap is a pure scheduling change."""
    args = (2, 4, 2, 0.01)
    overlap = run_distributed(_sp_drop_body, 3, args + (False,))[-1]
    blocking = run_distributed(_sp_drop_body, 3, args + (True,))[-1]
    for a, b in zip(overlap, blocking):
        assert abs(a - b) < 1e-6, (overlap, blocking)
