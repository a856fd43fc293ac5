"""Rank-math unit tests (no process group needed) + gloo group wiring.

Covers the reference topology semantics:
mp_size formula (comm.py:62-67), split_rank mapping (comm.py:139-152),
GEMS inversion (comm.py:77-80).
"""

import pytest

from dist_util import run_distributed
from mpi4dl_amd.comm import Communicator, compute_mp_size, normalize_spatial_parts


def test_mp_size_pure_lp():
    assert compute_mp_size(split_size=4) == 4
    assert compute_mp_size(split_size=1) == 1


def test_mp_size_spatial():
    # 4 tiles on first partition + 3 plain LP partitions = 4 + 3 = 7 ranks
    assert compute_mp_size(4, num_spatial_parts=4, spatial_size=1) == 7
    # two spatial partitions 4,2 + 2 LP = 4+2+2 = 8
    assert compute_mp_size(4, num_spatial_parts=[4, 2], spatial_size=2) == 8


def test_mp_size_local_dp():
    # spatial 4 tiles, 1 spatial partition, split 2, LOCAL_DP_LP=4:
    # 4 tiles + 1 LP partition x 4 local-DP ranks = 8
    assert compute_mp_size(2, num_spatial_parts=4, spatial_size=1, local_dp_lp=4) == 8


def test_normalize_spatial_parts():
    assert normalize_spatial_parts(4, 2) == [4, 4]
    assert normalize_spatial_parts([4, 2], 2) == [4, 2]
    assert normalize_spatial_parts([4], 2) == [4, 4]
    assert normalize_spatial_parts(4, 0) == []


def _mk_comm(world, rank, **kw):
    """Build a Communicator without init (pure rank math)."""
    import mpi4dl_amd.comm as C

    class FakeDist:
        @staticmethod
        def is_initialized():
            return True

        @staticmethod
        def get_rank():
            return rank

        @staticmethod
        def get_world_size():
            return world

        @staticmethod
        def new_group(ranks=None):
            return tuple(ranks)

        @staticmethod
        def get_world_size_group(group):
            return len(group)

    return FakeDist


def test_split_rank_mapping(monkeypatch):
    import mpi4dl_amd.comm as C

    fake = _mk_comm(7, 5)
    monkeypatch.setattr(C, "dist", fake)
    comm = Communicator(
        split_size=4,
        ENABLE_SPATIAL=True,
        num_spatial_parts=4,
        spatial_size=1,
        DISABLE_INIT=True,
    )
    # ranks 0-3 are tiles of partition 0; 4,5,6 are partitions 1,2,3
    assert [comm.get_split_rank(r) for r in range(7)] == [0, 0, 0, 0, 1, 2, 3]
    assert comm.ranks_of_partition(0) == [0, 1, 2, 3]
    assert comm.ranks_of_partition(2) == [5]
    assert comm.split_rank == 2  # rank 5


def test_split_rank_skewed(monkeypatch):
    import mpi4dl_amd.comm as C

    monkeypatch.setattr(C, "dist", _mk_comm(8, 0))
    comm = Communicator(
        split_size=4,
        ENABLE_SPATIAL=True,
        num_spatial_parts=[4, 2],
        spatial_size=2,
        DISABLE_INIT=True,
    )
    assert [comm.get_split_rank(r) for r in range(8)] == [0, 0, 0, 0, 1, 1, 2, 3]


def test_gems_inversion(monkeypatch):
    import mpi4dl_amd.comm as C

    monkeypatch.setattr(C, "dist", _mk_comm(4, 1))
    comm = Communicator(split_size=4, ENABLE_MASTER=True, DISABLE_INIT=True)
    assert comm.local_rank == 2  # mp_size-1-1
    # engine peer map: inverse engine position j lives on global mp-1-j
    assert comm.engine_peer(1, gems_inverse=True) == 2
    assert comm.engine_peer(1, gems_inverse=False) == 1


def test_local_dp_split_rank(monkeypatch):
    import mpi4dl_amd.comm as C

    monkeypatch.setattr(C, "dist", _mk_comm(8, 0))
    comm = Communicator(
        split_size=2,
        ENABLE_SPATIAL=True,
        num_spatial_parts=4,
        spatial_size=1,
        LOCAL_DP_LP=4,
        DISABLE_INIT=True,
    )
    assert [comm.get_split_rank(r) for r in range(8)] == [0, 0, 0, 0, 1, 1, 1, 1]
    assert comm.ranks_of_partition(1) == [4, 5, 6, 7]


def _selftest_body(rank, world):
    comm = Communicator(split_size=2, DISABLE_INIT=False, backend="gloo")
    comm.self_test()
    return comm.describe()


def _outer_dp_body(rank, world):
    # world 4 = 2 replicas x mp_size 2
    comm = Communicator(split_size=2, backend="gloo")
    assert comm.dp_size == 2
    assert comm.replica == rank // 2
    assert comm.outer_dp_group is not None
    return True


def test_gloo_groups_and_selftest():
    from dist_util import run_distributed

    out = run_distributed(_selftest_body, world_size=2)
    assert len(out) == 2


def test_outer_dp_groups_gloo():
    from dist_util import run_distributed

    run_distributed(_outer_dp_body, world_size=4)


def _fp16_allreduce_body(rank, world):
    import torch

    from mpi4dl_amd.comm import Communicator, GradReducer

    comm = Communicator(split_size=1, backend="gloo")  # world ranks = DP replicas
    torch.manual_seed(rank)
    m = torch.nn.Linear(8, 4)
    (m(torch.randn(2, 8)).sum()).backward()
    comp = GradReducer(comm, fp16_allreduce=True)
    ref = [p.grad.clone() for p in m.parameters()]
    comp.apply_allreduce(m)
    got = [p.grad.clone() for p in m.parameters()]
    # reconstruct expectation: bf16-rounded local grads averaged in bf16
    return [g.tolist() for g in got], [r.tolist() for r in ref]


def test_fp16_allreduce_close():
    """--fp16-allreduce reduces a bf16 copy: result must equal the fp32
    average within bf16 rounding (~3 decimal digits)."""
    import torch

    got = run_distributed(_fp16_allreduce_body, 2)
    g0 = [torch.tensor(t) for t in got[0][0]]
    g1 = [torch.tensor(t) for t in got[1][0]]
    # both ranks converge to the same reduced gradient
    for a, b in zip(g0, g1):
        assert torch.allclose(a, b)
    # and it is close to the true fp32 mean of the two local grads
    r0 = [torch.tensor(t) for t in got[0][1]]
    r1 = [torch.tensor(t) for t in got[1][1]]
    for a, x, y in zip(g0, r0, r1):
        assert torch.allclose(a, (x + y) / 2, rtol=2e-2, atol=2e-2)
