"""Checkpoint/resume tests: shard round-trip, trajectory resume parity,
and consolidate-to-single-model."""

import os
import tempfile

import torch
import torch.nn as nn

from dist_util import run_distributed

IMG = 32
NCLS = 10


def _ckpt_body(rank, world, tmpdir):
    from mpi4dl_amd.checkpoint import load_checkpoint, save_checkpoint
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    torch.manual_seed(0)
    model = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    gen = model_generator(model, world, input_size=(2, 3, IMG, IMG))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=0.01, momentum=0.9)
    eng = train_model(gen, comm.local_rank, 2, 1, comm, optimizer=opt,
                      device=torch.device("cpu"))

    torch.manual_seed(42)
    data = [(torch.randn(2, 3, IMG, IMG), torch.randint(0, NCLS, (2,)))
            for _ in range(4)]

    # train 2 steps, checkpoint, train 2 more -> loss trajectory A
    for x, y in data[:2]:
        eng.run_step(x, y)
        eng.update()
    save_checkpoint(tmpdir, gen.models, opt, comm, balance=gen.balance,
                    extra={"step": 2})
    lossesA = []
    for x, y in data[2:]:
        l, _, _ = eng.run_step(x, y)
        eng.update()
        lossesA.append(l)

    # rebuild fresh, restore, replay steps 3-4 -> must match
    torch.manual_seed(123)  # different init on purpose
    model_b = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    gen_b = model_generator(model_b, world, input_size=(2, 3, IMG, IMG))
    gen_b.get_output_shapes()
    gen_b.ready_model(comm.local_rank, device=torch.device("cpu"))
    opt_b = torch.optim.SGD(gen_b.models.parameters(), lr=0.01, momentum=0.9)
    eng_b = train_model(gen_b, comm.local_rank, 2, 1, comm, optimizer=opt_b,
                        device=torch.device("cpu"))
    extra = load_checkpoint(tmpdir, gen_b.models, opt_b, comm)
    assert extra["step"] == 2
    lossesB = []
    for x, y in data[2:]:
        l, _, _ = eng_b.run_step(x, y)
        eng_b.update()
        lossesB.append(l)
    for a, b in zip(lossesA, lossesB):
        assert abs(a - b) < 1e-6, (lossesA, lossesB)
    return lossesA


def test_checkpoint_resume_parity(tmp_path):
    run_distributed(_ckpt_body, 2, (str(tmp_path),))


def _consolidate_body(rank, world, tmpdir):
    from mpi4dl_amd.checkpoint import save_checkpoint
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator

    comm = Communicator(split_size=world, backend="gloo")
    torch.manual_seed(0)
    model = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    gen = model_generator(model, world, input_size=(2, 3, IMG, IMG))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    save_checkpoint(tmpdir, gen.models, None, comm, balance=gen.balance)
    return True


def test_consolidate(tmp_path):
    from mpi4dl_amd.checkpoint import consolidate
    from mpi4dl_amd.models.resnet import get_resnet_v1

    run_distributed(_consolidate_body, 3, (str(tmp_path),))
    full = consolidate(str(tmp_path))
    torch.manual_seed(0)
    ref = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    ref_sd = ref.state_dict()
    assert set(full.keys()) == set(ref_sd.keys()), (
        sorted(set(ref_sd) - set(full))[:5],
        sorted(set(full) - set(ref_sd))[:5],
    )
    for k in ref_sd:
        assert torch.equal(full[k], ref_sd[k]), k


def _reshard_body(rank, world, tmpdir):
    """Resume shards saved at split=3 into a split=2 run via consolidate."""
    from mpi4dl_amd.checkpoint import consolidate, load_from_consolidated
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=world, backend="gloo")
    full = consolidate(tmpdir)  # written by the 3-way run
    torch.manual_seed(99)  # deliberately different init
    model = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    gen = model_generator(model, world, input_size=(2, 3, IMG, IMG))
    gen.get_output_shapes()
    gen.ready_model(comm.local_rank, device=torch.device("cpu"))
    load_from_consolidated(full, gen.models)
    eng = train_model(gen, comm.local_rank, 2, 1, comm, device=torch.device("cpu"))
    torch.manual_seed(5)
    x = torch.randn(2, 3, IMG, IMG)
    y = torch.randint(0, NCLS, (2,))
    loss, _, _ = eng.run_eval(x, y)
    if rank == world - 1:
        # reference: the consolidated full model on one process
        ref = get_resnet_v1((2, 3, IMG, IMG), NCLS, n=1, num_filters=8)
        ref.load_state_dict(full)
        ref.eval()
        with torch.no_grad():
            rl = torch.nn.functional.cross_entropy(ref(x), y)
        assert abs(loss - float(rl)) < 1e-4, (loss, float(rl))
    return True


def test_reshard_via_consolidate(tmp_path):
    run_distributed(_consolidate_body, 3, (str(tmp_path),))
    run_distributed(_reshard_body, 2, (str(tmp_path),))


def _gems_ckpt_body(rank, world, tmpdir, phase):
    """phase 'save': train 1 step, checkpoint. phase 'resume': restore,
    verify BOTH replica engines hold the checkpointed weights."""
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from runner import make_engines

    from mpi4dl_amd import checkpoint as ckpt
    from mpi4dl_amd.parser import get_parser

    args = get_parser().parse_args([
        "--model", "resnet", "--batch-size", "2", "--parts", "1",
        "--split-size", "2", "--image-size", "32", "--num-layers", "9",
        "--num-filters", "4", "--num-classes", "10", "--backend", "gloo",
    ])
    step, comm, extras = make_engines(args, "gems")
    eng = extras["engine"]
    if phase == "save":
        torch.manual_seed(42)
        x = torch.randn(4, 3, 32, 32)  # 2x batch for the two replicas
        y = torch.randint(0, 10, (4,))
        step(x, y)
        ckpt.save_checkpoint(tmpdir, extras["gen"].models, None, comm)
        # numpy: torch tensors ride the queue via shm FDs that die with
        # the child process
        return [p.detach().numpy().copy()
                for p in eng.train_model1.models.parameters()]
    # resume path (fresh random init on purpose: different seed)
    ckpt.load_checkpoint(tmpdir, extras["gen"].models, None, comm)
    eng.sync_models()
    p1 = [p.detach().numpy().copy()
          for p in eng.train_model1.models.parameters()]
    p2 = [p.detach().numpy().copy()
          for p in eng.train_model2.models.parameters()]
    return p1, p2


def _gems_save(rank, world, tmpdir):
    return _gems_ckpt_body(rank, world, tmpdir, "save")


def _gems_resume(rank, world, tmpdir):
    return _gems_ckpt_body(rank, world, tmpdir, "resume")


def test_gems_resume_restores_both_replicas(tmp_path):
    saved = run_distributed(_gems_save, 2, (str(tmp_path),))
    out = run_distributed(_gems_resume, 2, (str(tmp_path),))
    for rank in range(2):
        p1, p2 = out[rank]
        # replica 1 == what rank `rank` checkpointed
        for a, b in zip(p1, saved[rank]):
            assert (a == b).all()
        # replica 2 on rank r == replica 1's stage from the mirror rank
        mirror = 1 - rank
        for a, b in zip(p2, saved[mirror]):
            assert (a == b).all()
