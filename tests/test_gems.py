"""GEMS bidirectional-parallelism parity tests (gloo).

GEMS with paired gradient averaging is mathematically data parallelism
over the doubled batch — the distributed trajectory must match a serial
model trained on the full 2B batch."""

import torch
import torch.nn as nn

from dist_util import run_distributed

IMG = 32
NCLS = 10


def _build():
    from mpi4dl_amd.models.resnet import get_resnet_v1

    torch.manual_seed(0)
    return get_resnet_v1((2, 3, IMG, IMG), num_classes=NCLS, n=1, num_filters=8)


def _serial_losses(steps, B, parts, lr):
    """Ground truth: one model, 2B batch, grads averaged over the two
    halves exactly as GEMS pairs them."""
    model = _build()
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(2 * B, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (2 * B,))
        half_losses = []
        for h in range(2):
            xh, yh = x[h * B : (h + 1) * B], y[h * B : (h + 1) * B]
            total = 0.0
            for px, py in zip(xh.chunk(parts), yh.chunk(parts)):
                loss = crit(model(px).float(), py)
                (loss / (parts * 2)).backward()
                total += float(loss.detach())
            half_losses.append(total / parts)
        opt.step()
        opt.zero_grad(set_to_none=False)
        losses.append(sum(half_losses) / 2)
    return losses


def _gems_body(rank, world, steps, B, parts, lr, comm_opt):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.parallel.gems import train_model_master
    from mpi4dl_amd.parallel.partition import model_generator

    comm = Communicator(split_size=world, ENABLE_GEMS=True, backend="gloo")
    r = comm.rank % comm.mp_size

    def mkgen(pos):
        model = _build()
        gen = model_generator(model, world, input_size=(B // parts, 3, IMG, IMG))
        gen.get_output_shapes()
        gen.ready_model(pos, device=torch.device("cpu"))
        return gen

    gen1 = mkgen(r)
    gen2 = mkgen(comm.mp_size - 1 - r)
    eng = train_model_master(
        gen1, gen2, B, parts, comm, enable_comm_opt=comm_opt, lr=lr
    )
    eng.sync_models()
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(2 * B, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (2 * B,))
        loss, _, _ = eng.run_step(x, y)
        if comm_opt and not eng._swap.local:
            # MASTER-OPT: leg A (replica 1's flat grads) must already be
            # in flight here — issued mid-run_step, before replica 2's
            # step, so it overlaps that compute on RCCL
            assert eng._swap._leg_a is not None, "leg A not issued mid-step"
        eng.allreduce_and_update()
        losses.append(loss)
    # return per-engine losses: loss only meaningful where last stage lives
    return losses


def test_gems_lp_parity():
    steps, B, parts, lr = 3, 2, 1, 0.01
    expected = _serial_losses(steps, B, parts, lr)
    got = run_distributed(_gems_body, 2, (steps, B, parts, lr, False))
    # engine1's last stage on rank 1 reports loss1; engine2's on rank 0
    # reports loss2; each rank's run_step returns (l1+l2)/2 but only the
    # rank hosting each last stage sees a nonzero piece — combine:
    combined = [a + b for a, b in zip(got[0], got[1])]
    for e, g in zip(expected, combined):
        assert abs(e - g) < 2e-4, (expected, combined)


def test_gems_lp_parity_comm_opt():
    steps, B, parts, lr = 3, 2, 1, 0.01
    expected = _serial_losses(steps, B, parts, lr)
    got = run_distributed(_gems_body, 2, (steps, B, parts, lr, True))
    combined = [a + b for a, b in zip(got[0], got[1])]
    for e, g in zip(expected, combined):
        assert abs(e - g) < 2e-4, (expected, combined)


def _gems_sp_body(rank, world, steps, B, parts, lr, comm_opt=False):
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.gems import train_spatial_model_master
    from mpi4dl_amd.parallel.partition import model_generator

    split, nsp = 3, 2
    comm = Communicator(
        split_size=split,
        ENABLE_SPATIAL=True,
        num_spatial_parts=nsp,
        spatial_size=1,
        ENABLE_GEMS=True,
        backend="gloo",
    )
    r = comm.rank % comm.mp_size
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    ncells = len(probe)
    balance = [2, 2, ncells - 4]

    def mkgen(pos, inverse):
        plan = SpatialPlan(comm, balance, "vertical", gems_inverse=inverse)
        torch.manual_seed(0)
        model = resnet_spatial.get_resnet_v1(
            (B // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
        )
        gen = model_generator(
            model, split, input_size=(B // parts, 3, IMG, IMG), balance=balance
        )
        gen.get_output_shapes()
        gen.ready_model(comm.get_split_rank(pos), device=torch.device("cpu"))
        return gen

    gen1 = mkgen(r, False)
    gen2 = mkgen(comm.mp_size - 1 - r, True)
    eng = train_spatial_model_master(
        gen1, gen2, B, parts, comm, slice_method="vertical", lr=lr,
        enable_comm_opt=comm_opt,
    )
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(2 * B, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (2 * B,))
        loss, _, _ = eng.run_step(x, y)
        if comm_opt and not eng._swap.local:
            assert eng._swap._leg_a is not None, "leg A not issued mid-step"
        eng.allreduce_and_update()
        losses.append(loss)
    return losses


def test_gems_sp_parity():
    steps, B, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, B, parts, lr)
    # split 3, 2 tiles -> mp = 4; engine2 tiles on mirrored ranks 3,2
    got = run_distributed(_gems_sp_body, 4, (steps, B, parts, lr))
    combined = [sum(g[s] for g in got) for s in range(steps)]
    for e, g in zip(expected, combined):
        assert abs(e - g) < 2e-4, (expected, combined)


def test_gems_sp_parity_comm_opt():
    """MASTER-OPT in the SPATIAL master: the flat-grad mirror swap (now
    actually taken when --enable-master-comm-opt) must produce the same
    trajectory as the pair allreduce."""
    steps, B, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, B, parts, lr)
    got = run_distributed(_gems_sp_body, 4, (steps, B, parts, lr, True))
    combined = [sum(g[s] for g in got) for s in range(steps)]
    for e, g in zip(expected, combined):
        assert abs(e - g) < 2e-4, (expected, combined)


def test_gems_lp_parity_odd_stages():
    """GEMS with an ODD stage count (mp=3): the middle rank hosts both
    replicas' copy of its stage and must average the pair locally (a
    {r, r} process group is invalid). Trajectory == DP over 2x batch."""
    steps, B, parts, lr = 2, 2, 1, 0.01
    expected = _serial_losses(steps, B, parts, lr)
    got = run_distributed(_gems_body, 3, (steps, B, parts, lr, False))
    combined = [sum(g[s] for g in got) for s in range(steps)]
    for e, g in zip(expected, combined):
        assert abs(e - g) < 2e-4, (expected, combined)
