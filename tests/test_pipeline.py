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


def _dp_body(rank, world, steps, batch, parts, lr):
    """Outer DP x LP: world 4 = 2 replicas x 2 stages; each replica
    trains half the global batch; grads averaged across replicas."""
    import torch.distributed as dist

    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=2, backend="gloo")
    model = _build()
    gen = model_generator(model, 2, input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(gen, comm.local_rank, batch, parts, comm, optimizer=opt,
                      device=torch.device("cpu"))
    red = GradReducer(comm)
    xs, ys = _data(steps, 2 * batch)  # global batch; my replica takes half
    lo = comm.replica * batch
    losses = []
    for x, y in zip(xs, ys):
        loss, _, _ = eng.run_step(x[lo : lo + batch], y[lo : lo + batch])
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    # the two replicas' weights must be identical after sync steps
    flat = torch.cat([p.reshape(-1) for p in eng.models.parameters()])
    peer = (comm.rank + comm.mp_size) % comm.world_size
    other = torch.empty_like(flat)
    from mpi4dl_amd import p2p

    if comm.rank < peer:
        p2p.send_tensors([flat], peer, tag_base=9000)
        p2p.recv_tensors([other], peer, tag_base=9001)
    else:
        p2p.recv_tensors([other], peer, tag_base=9000)
        p2p.send_tensors([flat], peer, tag_base=9001)
    assert torch.allclose(flat, other, atol=1e-6)
    return losses


def test_outer_dp_parity():
    steps, batch, parts, lr = 2, 4, 2, 0.01
    # serial ground truth: batch 2B with grads = mean of the two halves
    expected = _serial_losses(steps, 2 * batch, 2 * parts, lr)
    got = run_distributed(_dp_body, 4, (steps, batch, parts, lr))
    # loss on each replica's last stage covers its half; mean over the two
    # replicas' losses equals the serial full-batch loss
    for e, g0, g1 in zip(expected, got[1], got[3]):
        assert abs(e - (g0 + g1) / 2) < 2e-4, (expected, got[1], got[3])


def _eval_body(rank, world):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, world, input_size=(2, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    eng = train_model(gen, comm.local_rank, 4, 2, comm, device=torch.device("cpu"))
    torch.manual_seed(5)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, 10, (4,))
    loss, corr, seen = eng.run_eval(x, y)
    if rank == world - 1:
        model.eval()
        with torch.no_grad():
            ref = torch.nn.functional.cross_entropy(model(x), y)
        assert abs(loss - float(ref)) < 1e-4, (loss, float(ref))
        assert seen == 4
    return True


def test_run_eval_matches_serial():
    run_distributed(_eval_body, 2, ())


def _dp_overlap_body(rank, world, steps, batch, parts, lr):
    """Same as _dp_body but gradients reduce via the bucketed overlap
    hooks during backward instead of one post-step allreduce."""
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=2, backend="gloo")
    model = _build()
    gen = model_generator(model, 2, input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(gen, comm.local_rank, batch, parts, comm, optimizer=opt,
                      device=torch.device("cpu"))
    red = GradReducer(comm)
    red.prepare_overlap(gen.models, comm.outer_dp_group, bucket_mb=0.01)
    xs, ys = _data(steps, 2 * batch)
    lo = comm.replica * batch
    losses = []
    for x, y in zip(xs, ys):
        loss, _, _ = eng.run_step(x[lo : lo + batch], y[lo : lo + batch])
        red.finish_overlap(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def test_outer_dp_bucketed_overlap_parity():
    steps, batch, parts, lr = 2, 4, 2, 0.01
    expected = _serial_losses(steps, 2 * batch, 2 * parts, lr)
    got = run_distributed(_dp_overlap_body, 4, (steps, batch, parts, lr))
    for e, g0, g1 in zip(expected, got[1], got[3]):
        assert abs(e - (g0 + g1) / 2) < 2e-4, (expected, got[1], got[3])


def _bf16_act_body(rank, world, steps, batch, parts, lr):
    """bf16 activation boundaries (the GPU message dtype) on gloo: the
    pipeline must converge within bf16 tolerance of the fp32 run."""
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, world, input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(
        gen, comm.local_rank, batch, parts, comm, optimizer=opt,
        device=torch.device("cpu"), act_dtype=torch.bfloat16,
    )
    xs, ys = _data(steps, batch)
    losses = []
    for x, y in zip(xs, ys):
        loss, _, _ = eng.run_step(x, y)
        eng.update()
        losses.append(loss)
    return losses


def test_bf16_activation_boundaries():
    steps, batch, parts, lr = 3, 4, 2, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_bf16_act_body, 2, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 0.05, (expected, got)  # bf16 message tolerance


def _pipeline_1f1b_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    model = _build()
    gen = model_generator(model, world, input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(gen, comm.local_rank, batch, parts, comm, optimizer=opt,
                      device=torch.device("cpu"), schedule="1f1b")
    xs, ys = _data(steps, batch)
    losses = []
    for x, y in zip(xs, ys):
        loss, _, _ = eng.run_step(x, y)
        eng.update()
        losses.append(loss)
    return losses


def test_1f1b_schedule_parity():
    steps, batch, parts, lr = 2, 8, 4, 0.01
    expected = _serial_losses(steps, batch, parts, lr)
    got = run_distributed(_pipeline_1f1b_body, 3, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 1e-4, (expected, got)


def _act_ckpt_body(rank, world, steps, batch, parts, lr):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    torch.manual_seed(0)
    model = get_resnet_v1((batch // parts, 3, 32, 32), 10, n=1, num_filters=8)
    gen = model_generator(model, world,
                          input_size=(batch // parts, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model(gen, comm.local_rank, batch, parts, comm, optimizer=opt,
                      device=torch.device("cpu"), act_ckpt=True)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, 32, 32)
        y = torch.randint(0, 10, (batch,))
        loss, _, _ = eng.run_step(x, y)
        eng.update()
        losses.append(loss)
    return losses


def test_act_ckpt_parity():
    """Activation checkpointing (recompute in backward) must not change
    the training trajectory — same losses as plain serial training."""
    from mpi4dl_amd.models.resnet import get_resnet_v1

    steps, batch, parts, lr = 3, 4, 2, 0.01
    torch.manual_seed(0)
    model = get_resnet_v1((batch // parts, 3, 32, 32), 10, n=1, num_filters=8)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    torch.manual_seed(42)
    expected = []
    for _ in range(steps):
        x = torch.randn(batch, 3, 32, 32)
        y = torch.randint(0, 10, (batch,))
        total = 0.0
        for px, py in zip(x.chunk(parts), y.chunk(parts)):
            loss = crit(model(px).float(), py)
            (loss / parts).backward()
            total += float(loss.detach())
        opt.step()
        opt.zero_grad(set_to_none=False)
        expected.append(total / parts)
    got = run_distributed(_act_ckpt_body, 2, (steps, batch, parts, lr))[-1]
    for e, g in zip(expected, got):
        assert abs(e - g) < 2e-4, (expected, got)
