"""AmoebaNet-D: structure, LP pipeline parity (tuple activations), and
SP parity with the tuple-carrying spatial seam."""

import torch
import torch.nn as nn

from dist_util import run_distributed

IMG = 64
NCLS = 10
LAYERS = 3
FILTERS = 32


def _build(plan=None):
    from mpi4dl_amd.models.amoebanet import amoebanetd

    torch.manual_seed(0)
    return amoebanetd(NCLS, LAYERS, FILTERS, plan=plan)


def test_amoebanet_forward_backward():
    m = _build()
    x = torch.randn(2, 3, IMG, IMG)
    y = m(x)
    assert y.shape == (2, NCLS)
    y.sum().backward()


def test_amoebanet_meta_shapes():
    from mpi4dl_amd.parallel.partition import model_generator

    m = _build()
    gen = model_generator(m, 3, input_size=(2, 3, IMG, IMG))
    shapes = gen.get_output_shapes()
    assert len(shapes) == 3
    # intermediate stages carry (x, skip) tuples
    assert isinstance(shapes[0], list) and len(shapes[0]) == 2


def _serial_losses(steps, batch, parts, lr):
    model = _build()
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


def _lp_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, world, input_size=(batch // parts, 3, IMG, IMG))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(
        gen, comm.local_rank, batch, parts, comm, optimizer=opt,
        device=torch.device("cpu"),
    )
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        eng.update()
        losses.append(loss)
    return losses


def test_amoebanet_lp_parity():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_lp_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def _sp_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models.amoebanet import amoebanetd
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
    probe = _build()
    ncells = len(probe)
    balance = [4, ncells - 4]
    plan = SpatialPlan(comm, balance, "vertical")
    torch.manual_seed(0)
    from mpi4dl_amd.models.amoebanet import amoebanetd as build

    model = build(NCLS, LAYERS, FILTERS, plan=plan)
    gen = model_generator(
        model, split, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm,
        slice_method="vertical", optimizer=opt, device=torch.device("cpu"),
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


def test_amoebanet_sp_parity():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # 2 tiles + 1 LP rank = 3 ranks; tuple (x, skip) crosses the joint seam
    got = run_distributed(_sp_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)


def _sp_square_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models.amoebanet import amoebanetd as build
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    split = 2
    comm = Communicator(
        split_size=split, ENABLE_SPATIAL=True, num_spatial_parts=4,
        spatial_size=1, backend="gloo",
    )
    probe = _build()
    balance = [4, len(probe) - 4]
    plan = SpatialPlan(comm, balance, "square")
    torch.manual_seed(0)
    model = build(NCLS, LAYERS, FILTERS, plan=plan)
    gen = model_generator(
        model, split, input_size=(batch // parts, 3, IMG, IMG), balance=balance
    )
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm,
        slice_method="square", optimizer=opt, device=torch.device("cpu"),
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


def test_amoebanet_sp_square_parity():
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    # 4 square tiles (8-neighbour halos incl. corners) + 1 LP rank
    got = run_distributed(_sp_square_body, 5, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 1e-3, (expected, got)  # fp32 reduction-order noise


def _lp_ckpt_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, world, input_size=(batch // parts, 3, IMG, IMG))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(
        gen, comm.local_rank, batch, parts, comm, optimizer=opt,
        device=torch.device("cpu"), act_ckpt=True,
    )
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        eng.update()
        losses.append(loss)
    return losses


def test_amoebanet_act_ckpt_parity():
    """Tuple (x, skip) activations through per-cell non-reentrant
    checkpoint: trajectory must match serial."""
    steps, batch, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_lp_ckpt_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)
