"""LP+PP engine parity test: 2-stage pipeline on gloo must reproduce the
single-process training trajectory bit-closely."""

import torch
import torch.nn as nn

from dist_util import run_distributed


def _build(seed=0):
    from mpi4dl_amd.models.resnet import get_resnet_v1

    torch.manual_seed(seed)
    return get_resnet_v1((4, 3, 32, 32), num_classes=10, n=1, num_filters=8)


def _data(steps=3, batch=4):
    torch.manual_seed(42)
    xs = [torch.randn(batch, 3, 32, 32) for _ in range(steps)]
    ys = [torch.randint(0, 10, (batch,)) for _ in range(steps)]
    return xs, ys


def _serial_losses(steps=3, batch=4, parts=2, lr=0.01):
    """Ground truth: same math on one process."""
    model = _build()
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    xs, ys = _data(steps, batch)
    losses = []
    for x, y in zip(xs, ys):
        total = 0.0
        for px, py in zip(x.chunk(parts), y.chunk(parts)):
            loss = crit(model(px).float(), py)
            (loss / parts).backward()
            total += float(loss.detach())
        opt.step()
        opt.zero_grad(set_to_none=False)
        losses.append(total / parts)
    return losses


def _pipeline_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, split_size=world, input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    engine = train_model(
        gen,
        local_rank=comm.local_rank,
        batch_size=batch,
        parts=parts,
        comm=comm,
        optimizer=opt,
        device=torch.device("cpu"),
    )
    xs, ys = _data(steps, batch)
    losses = []
    for x, y in zip(xs, ys):
        loss, _, _ = engine.run_step(x, y)
        engine.update()
        losses.append(loss)
    return losses


def test_lp_two_stage_parity():
    steps, batch, parts, lr = 3, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    results = run_distributed(_pipeline_body, 2, (steps, batch, parts, lr))
    got = results[1]  # last stage computes loss
    assert len(got) == steps
    for e, g in zip(expected, got):
        assert abs(e - g) < 1e-4, (expected, got)


def test_lp_three_stage_parity():
    steps, batch, parts, lr = 2, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    results = run_distributed(_pipeline_body, 3, (steps, batch, parts, lr))
    got = results[2]
    for e, g in zip(expected, got):
        assert abs(e - g) < 1e-4, (expected, got)


def test_single_rank_pipeline():
    """split_size=1 degenerates to plain local training (bench N=1 path)."""
    steps, batch, parts, lr = 2, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    results = run_distributed(_pipeline_body, 1, (steps, batch, parts, lr))
    for e, g in zip(expected, results[0]):
        assert abs(e - g) < 1e-4
