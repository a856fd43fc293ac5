"""Communication core: topology math, process groups, gradient sync.

MI355X-native replacement for the reference's MPI substrate
(reference: src/torchgems/comm.py — MPIComm :44-137, SyncAllreduce
:335-523, sync_comms_for_master :312-332). Differences by design:

* rendezvous via torch.distributed env:// (torchrun / bench driver), no
  MPI launcher; backend "nccl" IS RCCL on ROCm, "gloo" for CPU tests;
* one process per GPU, device = LOCAL_RANK (the reference hard-codes
  4 GPUs/node, comm.py:39 — we bind to LOCAL_RANK % device_count);
* no message tags (RCCL has none): per-purpose process groups + stream
  ordering replace the reference's tag discipline (spatial.py:170-175);
* gradient reduction uses a persistent flat buffer that the parameter
  .grads are *views into*, so there is no per-step flatten/unflatten
  (the reference re-``torch.cat``s every step — comm.py:414-438).

Topology vocabulary (identical semantics to the reference):

* ``split_size``     — number of LP (layer/pipeline) partitions.
* ``num_spatial_parts`` — tiles per spatial partition (int or list, one
  entry per spatial partition).
* ``spatial_size``   — how many leading partitions are spatial.
* ``local_dp_lp``    — LBANN-style data-parallel degree *inside* each LP
  partition that follows the spatial ones.
* ``mp_size``        — ranks in one model-parallel clique
  (comm.py:62-67):
  ``split_size + sum(spatial_parts) - spatial_size
  + (split_size - spatial_size) * (local_dp_lp - 1)``
* outer DP: world_size // mp_size replicas, rank r belongs to replica
  r // mp_size, with local (in-clique) rank r % mp_size.
* GEMS master: the second replica engine sees local rank
  ``mp_size - 1 - local_rank`` (comm.py:77-80).
"""

from __future__ import annotations

import datetime
import logging
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

log = logging.getLogger(__name__)

_DEFAULT_TIMEOUT_S = 300


# ---------------------------------------------------------------------------
# Process bootstrap
# ---------------------------------------------------------------------------


def init_distributed(
    backend: Optional[str] = None,
    timeout_s: int = _DEFAULT_TIMEOUT_S,
) -> int:
    """Initialise torch.distributed from the environment and bind the GPU.

    Returns the global rank. Safe to call when already initialised.
    Single-process use (no RANK in env) initialises a world of 1 so all
    topology code runs unchanged.
    """
    if dist.is_initialized():
        return dist.get_rank()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29611")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("LOCAL_RANK", os.environ["RANK"])

    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local = int(os.environ["LOCAL_RANK"]) % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(local)
    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
    )
    return dist.get_rank()


def backend_is_nccl() -> bool:
    return dist.is_initialized() and dist.get_backend() == "nccl"


def current_device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


# ---------------------------------------------------------------------------
# Topology
# ---------------------------------------------------------------------------


def normalize_spatial_parts(
    num_spatial_parts, spatial_size: int
) -> List[int]:
    """Per-spatial-partition tile counts as a list of length spatial_size."""
    if spatial_size <= 0:
        return []
    if isinstance(num_spatial_parts, int):
        return [num_spatial_parts] * spatial_size
    parts = list(num_spatial_parts)
    if len(parts) == 1:
        parts = parts * spatial_size
    assert len(parts) >= spatial_size, (
        f"num_spatial_parts {parts} shorter than spatial_size {spatial_size}"
    )
    return parts[:spatial_size]


def compute_mp_size(
    split_size: int,
    num_spatial_parts=1,
    spatial_size: int = 0,
    local_dp_lp: int = 1,
) -> int:
    """Ranks in one model clique (reference comm.py:62-67)."""
    parts = normalize_spatial_parts(num_spatial_parts, spatial_size)
    return (
        split_size
        + sum(parts)
        - spatial_size
        + (split_size - spatial_size) * (local_dp_lp - 1)
    )


class Communicator:
    """Rank topology + process-group factory (reference MPIComm, comm.py:44).

    Builds every group eagerly and in identical order on all ranks (a
    torch.distributed requirement). Each group has a *purpose*; with RCCL
    a distinct group is a distinct communicator, which replaces the
    reference's MPI tag discipline for disambiguating concurrent traffic.
    """

    def __init__(
        self,
        split_size: int = 1,
        ENABLE_MASTER: bool = False,
        ENABLE_SPATIAL: bool = False,
        num_spatial_parts=1,
        spatial_size: int = 0,
        LOCAL_DP_LP: int = 1,
        DISABLE_INIT: bool = False,
        backend: Optional[str] = None,
        ENABLE_GEMS: bool = False,
    ):
        # ENABLE_GEMS builds the GEMS pair/mirror groups WITHOUT the
        # local-rank inversion (our single-communicator GEMS design:
        # parallel/gems.py supplies mirrored positions itself; the
        # reference instead builds a second inverted MPIComm,
        # comm.py:77-80 + sync_comms_for_master).
        if not ENABLE_SPATIAL:
            spatial_size = 0
            num_spatial_parts = 1
        if not DISABLE_INIT:
            init_distributed(backend=backend)
        assert dist.is_initialized(), "torch.distributed must be initialised"

        self.ENABLE_MASTER = ENABLE_MASTER
        self.ENABLE_GEMS = ENABLE_GEMS or ENABLE_MASTER
        self.ENABLE_SPATIAL = ENABLE_SPATIAL
        self.split_size = split_size
        self.spatial_size = spatial_size
        self.num_spatial_parts = num_spatial_parts
        self.spatial_parts = normalize_spatial_parts(num_spatial_parts, spatial_size)
        self.total_spatial_ranks = sum(self.spatial_parts)
        self.LOCAL_DP_LP = LOCAL_DP_LP

        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self.mp_size = compute_mp_size(
            split_size, num_spatial_parts, spatial_size, LOCAL_DP_LP
        )
        assert self.world_size % self.mp_size == 0, (
            f"world_size {self.world_size} not divisible by mp_size {self.mp_size}"
        )
        self.dp_size = self.world_size // self.mp_size
        self.replica = self.rank // self.mp_size

        # Local (in-clique) rank; GEMS master engines see the mirror image
        # (reference comm.py:77-80).
        self.local_rank = self.rank % self.mp_size
        if ENABLE_MASTER:
            self.local_rank = self.mp_size - 1 - self.local_rank

        self.split_rank = self.get_split_rank(self.local_rank)

        # -- process groups (identical creation order on every rank) --------
        self.outer_dp_group = None          # grads across replicas (same position)
        self.spatial_allreduce_groups = {}  # partition idx -> group (my replica)
        self.all_spatial_groups = {}        # (replica, partition) -> group
        self.master_pair_groups = {}        # local pos j -> 2-rank {j, mp-1-j} group
        self.scatter_gather_groups = {}     # spatial tile rank -> SP<->LP seam group
        self.local_dp_groups = {}           # lp partition idx -> local-DP group
        self._build_groups()

        if not DISABLE_INIT:
            self.self_test()

    # -- rank math ----------------------------------------------------------

    def get_split_rank(self, local_rank: int) -> int:
        """Partition index of an in-clique rank (reference comm.py:139-152)."""
        assert 0 <= local_rank < self.mp_size
        if local_rank < self.total_spatial_ranks:
            acc = 0
            for i, p in enumerate(self.spatial_parts):
                acc += p
                if local_rank < acc:
                    return i
        lp_offset = local_rank - self.total_spatial_ranks
        return self.spatial_size + lp_offset // self.LOCAL_DP_LP

    def first_local_rank_of_partition(self, part: int) -> int:
        """Lowest in-clique rank belonging to partition ``part``."""
        if part < self.spatial_size:
            return sum(self.spatial_parts[:part])
        return self.total_spatial_ranks + (part - self.spatial_size) * self.LOCAL_DP_LP

    def ranks_of_partition(self, part: int) -> List[int]:
        start = self.first_local_rank_of_partition(part)
        if part < self.spatial_size:
            return list(range(start, start + self.spatial_parts[part]))
        return list(range(start, start + self.LOCAL_DP_LP))

    def global_rank(self, local_rank: int, replica: Optional[int] = None) -> int:
        """In-clique rank -> global rank, for this (or a given) DP replica."""
        if replica is None:
            replica = self.replica
        return replica * self.mp_size + local_rank

    def engine_peer(self, local_rank: int, gems_inverse: bool = False) -> int:
        """Global rank hosting in-clique position ``local_rank`` as seen by an
        engine; a GEMS-inverse engine's position j lives on global rank
        mp_size-1-j (reference mp_pipeline.py:244-248)."""
        if gems_inverse:
            local_rank = self.mp_size - 1 - local_rank
        return self.global_rank(local_rank)

    # -- group construction --------------------------------------------------

    def _new_group(self, ranks: Sequence[int]):
        ranks = sorted(set(ranks))
        if len(ranks) <= 1:
            return None
        return dist.new_group(ranks=ranks)

    def _build_groups(self):
        # Outer DP: same in-clique position across replicas (comm.py:161-168).
        if self.dp_size > 1:
            for j in range(self.mp_size):
                g = self._new_group(
                    [i * self.mp_size + j for i in range(self.dp_size)]
                )
                if j == self.rank % self.mp_size:
                    self.outer_dp_group = g

        # Spatial allreduce: tile ranks of each spatial partition, per replica
        # (comm.py:197-248). Also used for initial weight broadcast.
        for rep in range(self.dp_size):
            for part in range(self.spatial_size):
                ranks = [rep * self.mp_size + lr for lr in self.ranks_of_partition(part)]
                g = self._new_group(ranks)
                self.all_spatial_groups[(rep, part)] = g
                if rep == self.replica:
                    self.spatial_allreduce_groups[part] = g

        # GEMS master pairs: {j, mp-1-j} per replica (comm.py:170-195) — the
        # two global ranks that host the same partition in the two mirrored
        # engines, used to average the paired grads.
        if self.mp_size > 1 and self.ENABLE_GEMS:
            for rep in range(self.dp_size):
                for j in range(self.mp_size):
                    a, b = j, self.mp_size - 1 - j
                    if a > b:
                        continue
                    if a == b:
                        # odd mp_size: the middle rank hosts BOTH engines'
                        # copy of its stage — no group; the reducer
                        # averages the pair locally (group=None path)
                        continue
                    g = self._new_group([rep * self.mp_size + a, rep * self.mp_size + b])
                    if rep == self.replica:
                        self.master_pair_groups[a] = g
                        self.master_pair_groups[b] = g

        # Mirror spatial groups for the GEMS-inverse engine: partition p's
        # tiles of replica 2 live on global ranks mp-1-lr (used for its
        # spatial grad reduction and TileBatchNorm sync).
        self.mirror_spatial_groups = {}
        if self.ENABLE_GEMS and self.spatial_size > 0:
            for rep in range(self.dp_size):
                for part in range(self.spatial_size):
                    ranks = [
                        rep * self.mp_size + (self.mp_size - 1 - lr)
                        for lr in self.ranks_of_partition(part)
                    ]
                    g = self._new_group(ranks)
                    if rep == self.replica:
                        self.mirror_spatial_groups[part] = g

        # SP<->LP scatter/gather seam groups for LOCAL_DP_LP (comm.py:250-276):
        # one group per tile rank of the *last* spatial partition: that tile
        # rank + the local-DP ranks of the first LP partition.
        if self.LOCAL_DP_LP > 1 and self.spatial_size >= 1:
            last_sp = self.spatial_size - 1
            first_lp_ranks = self.ranks_of_partition(self.spatial_size)
            for rep in range(self.dp_size):
                for tile_lr in self.ranks_of_partition(last_sp):
                    ranks = [rep * self.mp_size + tile_lr] + [
                        rep * self.mp_size + lr for lr in first_lp_ranks
                    ]
                    g = self._new_group(ranks)
                    if rep == self.replica:
                        self.scatter_gather_groups[tile_lr] = g

        # Local-DP groups inside each LP partition (comm.py:278-294).
        if self.LOCAL_DP_LP > 1:
            for rep in range(self.dp_size):
                for part in range(self.spatial_size, self.split_size):
                    ranks = [rep * self.mp_size + lr for lr in self.ranks_of_partition(part)]
                    g = self._new_group(ranks)
                    if rep == self.replica:
                        self.local_dp_groups[part] = g

    # -- diagnostics ---------------------------------------------------------

    def self_test(self):
        """Init-time wiring check: tiny allreduce on world + my groups
        (reference comm.py:304-309)."""
        dev = current_device() if backend_is_nccl() else torch.device("cpu")
        t = torch.ones(8, device=dev)
        dist.all_reduce(t)
        assert int(t[0].item()) == self.world_size, "world allreduce mismatch"
        for g in self._my_groups():
            t = torch.ones(8, device=dev)
            dist.all_reduce(t, group=g)
            assert int(t[0].item()) == dist.get_world_size(group=g)

    def _my_groups(self):
        out = []
        if self.outer_dp_group is not None:
            out.append(self.outer_dp_group)
        sp = self.spatial_allreduce_groups.get(self.split_rank)
        if sp is not None and self.split_rank < self.spatial_size and (
            self.local_rank < self.total_spatial_ranks
        ):
            out.append(sp)
        g = self.master_pair_groups.get(self.local_rank)
        if g is not None:
            out.append(g)
        return out

    def describe(self) -> str:
        return (
            f"Communicator(rank={self.rank}/{self.world_size}, mp_size={self.mp_size}, "
            f"dp_size={self.dp_size}, local_rank={self.local_rank}, "
            f"split_rank={self.split_rank}, spatial_parts={self.spatial_parts}, "
            f"local_dp_lp={self.LOCAL_DP_LP}, master={self.ENABLE_MASTER})"
        )


def sync_comms_for_master(comm1: Communicator, comm2: Communicator):
    """Wire the GEMS pair of communicators together (reference comm.py:312-332).

    comm1 is the normal clique, comm2 the mirrored (ENABLE_MASTER) one built
    with DISABLE_INIT=True. The mirrored engine must use the *same* group
    objects for each spatial partition's allreduce, looked up by its mirrored
    local rank, so paired reductions land on matching communicators.
    """
    comm2.spatial_allreduce_groups = {}
    for part in range(comm2.spatial_size):
        # partition `part` of engine2 lives on mirrored global ranks; its tile
        # group was built by comm1's loop over (replica, partition) — reuse it.
        mirror_part = part  # same partition index, mirrored rank placement
        g = comm1.all_spatial_groups.get((comm1.replica, mirror_part))
        comm2.spatial_allreduce_groups[mirror_part] = g
    comm2.master_pair_groups = comm1.master_pair_groups


# ---------------------------------------------------------------------------
# Gradient / weight synchronisation
# ---------------------------------------------------------------------------


class FlatGrads:
    """Make every parameter's .grad a view into ONE flat buffer.

    Autograd then accumulates straight into the buffer: reduction is a
    single all_reduce with zero pack/unpack work per step (the reference
    re-concatenates all grads every step — comm.py:414-438).
    """

    def __init__(
        self,
        module: torch.nn.Module,
        dtype: Optional[torch.dtype] = None,
        pad_to: int = 0,
    ):
        self.params = [p for p in module.parameters() if p.requires_grad]
        if not self.params:
            self.buffer = torch.zeros(0)
            return
        dev = self.params[0].device
        dtype = dtype or self.params[0].dtype
        total = sum(p.numel() for p in self.params)
        total = max((total + 3) // 4 * 4, pad_to)  # float4 kernels
        self.buffer = torch.zeros(total, device=dev, dtype=dtype)
        offset = 0
        for p in self.params:
            n = p.numel()
            view = self.buffer[offset : offset + n].view_as(p)
            if p.grad is not None:
                view.copy_(p.grad)  # keep grads accumulated before attach
            p.grad = view
            offset += n

    def zero_(self):
        if self.buffer.numel():
            self.buffer.zero_()

    def rescale_(self, divide_by: float):
        if divide_by != 1.0 and self.buffer.numel():
            self.buffer.div_(divide_by)

    @classmethod
    def get(cls, module: torch.nn.Module, pad_to: int = 0) -> "FlatGrads":
        """One FlatGrads per module (shared by GradReducer and FusedSGD
        so collectives and the optimizer use the SAME buffer)."""
        fg = getattr(module, "_mpi4dl_flatgrads", None)
        if fg is None or (pad_to and fg.buffer.numel() < pad_to):
            fg = cls(module, pad_to=pad_to)
            module._mpi4dl_flatgrads = fg
        return fg


class GradReducer:
    """Gradient & weight sync engine (reference SyncAllreduce, comm.py:335).

    Handles: initial weight broadcast within a group, flat-grad allreduce
    over outer-DP / spatial / local-DP groups, and the GEMS master pairing
    where the two mirrored engines' grads are averaged in a deadlock-free
    rank-dependent order (comm.py:460-504).
    """

    def __init__(self, comm: Communicator, fp16_allreduce: bool = False):
        self.comm = comm
        self._flat = {}  # id(module) -> FlatGrads
        # --fp16-allreduce: reduce a bf16 copy of the flat grads (halves
        # bytes on the xGMI links; bf16 keeps fp32's exponent range so
        # gradient magnitudes survive — fp16 would overflow). Opt-in:
        # trades ~3 bits of mantissa in the reduced gradient.
        self.fp16_allreduce = fp16_allreduce

    # -- weights -------------------------------------------------------------

    @staticmethod
    def broadcast_module(module: torch.nn.Module, group, src_rank: int):
        """Broadcast parameters (one flat message) so replicas start equal
        (reference sync_model, comm.py:368-400 does per-param broadcasts)."""
        if group is None:
            return
        params = [p.data for p in module.parameters()]
        if not params:
            return
        flat = torch.cat([p.reshape(-1) for p in params])
        dist.broadcast(flat, src=src_rank, group=group)
        offset = 0
        for p in params:
            n = p.numel()
            p.copy_(flat[offset : offset + n].view_as(p))
            offset += n

    def sync_model_spatial(self, module: torch.nn.Module):
        """Equalise weights across my spatial tile group (comm.py:374-380)."""
        if self.comm.split_rank < self.comm.spatial_size:
            g = self.comm.spatial_allreduce_groups.get(self.comm.split_rank)
            if g is not None:
                src = self.comm.global_rank(
                    self.comm.first_local_rank_of_partition(self.comm.split_rank)
                )
                self.broadcast_module(module, g, src)

    def sync_model_outer_dp(self, module: torch.nn.Module):
        g = self.comm.outer_dp_group
        if g is not None:
            src = self.comm.rank % self.comm.mp_size  # replica 0's same position
            self.broadcast_module(module, g, src)

    # -- grads ---------------------------------------------------------------

    def flat(self, module: torch.nn.Module) -> FlatGrads:
        return FlatGrads.get(module)

    def allreduce_grads(self, module: torch.nn.Module, group, divide_by: float = None):
        """Average grads over ``group`` (no-op for group=None / size 1)."""
        from .utils import GLOBAL_TIMER

        fg = self.flat(module)
        if group is None or fg.buffer.numel() == 0:
            return
        n = dist.get_world_size(group=group)
        with GLOBAL_TIMER.phase("grad/allreduce"):
            if self.fp16_allreduce:
                comp = fg.buffer.to(torch.bfloat16)
                dist.all_reduce(comp, group=group)
                fg.buffer.copy_(comp)
            else:
                dist.all_reduce(fg.buffer, group=group)
        fg.rescale_(divide_by if divide_by is not None else float(n))

    def apply_allreduce(self, module: torch.nn.Module, skip_group=None):
        """Standard path: spatial group first (tile replicas), then outer DP
        (reference apply_allreduce, comm.py:506-514).

        Scaling: tiles PARTITION the pixels of one batch, so the spatial
        group SUMS (each tile holds a partial weight grad). Local-DP and
        outer-DP groups each see the full weight grad of a batch shard,
        so those AVERAGE. This reproduces the serial gradient exactly
        (the reference folds the same arithmetic into divide_bs,
        comm.py:440-458).

        ``skip_group``: a group already reduced by the bucketed-overlap
        hooks (finish_overlap) — only the remaining groups run here."""
        if self.comm.ENABLE_SPATIAL and self.comm.split_rank < self.comm.spatial_size:
            g = self.comm.spatial_allreduce_groups.get(self.comm.split_rank)
            if g is not skip_group:
                self.allreduce_grads(module, g, divide_by=1.0)
        if self.comm.LOCAL_DP_LP > 1 and self.comm.split_rank >= self.comm.spatial_size:
            g = self.comm.local_dp_groups.get(self.comm.split_rank)
            if g is not skip_group:
                self.allreduce_grads(module, g)
        if self.comm.outer_dp_group is not skip_group:
            self.allreduce_grads(module, self.comm.outer_dp_group)

    def primary_reduce_group(self):
        """The innermost (largest-traffic) gradient group for this rank —
        the one worth overlapping with backward: the spatial tile group
        on tile ranks, the local-DP group on LP ranks under LOCAL_DP_LP,
        else the outer-DP group. Returns (group, divide_by)."""
        comm = self.comm
        if comm.ENABLE_SPATIAL and comm.split_rank < comm.spatial_size:
            return comm.spatial_allreduce_groups.get(comm.split_rank), 1.0
        if comm.LOCAL_DP_LP > 1 and comm.split_rank >= comm.spatial_size:
            return comm.local_dp_groups.get(comm.split_rank), None
        return comm.outer_dp_group, None

    def setup_overlap(self, module: torch.nn.Module, bucket_mb: float = 25.0):
        """Default production wiring (reference DDP wrap,
        mp_pipeline.py:92-124): register bucketed overlap on the primary
        group. Returns the overlap state (None when no group exists).
        The engine's run_step arms it on the last micro-batch; call
        ``finish_overlap`` then ``apply_allreduce(skip_group=...)``."""
        group, divide = self.primary_reduce_group()
        if group is None:
            return None
        prev = getattr(module, "_mpi4dl_overlap", None)
        if prev is not None and prev["group"] is group:
            return prev  # hooks already registered for this group
        state = self.prepare_overlap(module, group, divide_by=divide,
                                     bucket_mb=bucket_mb)
        return state

    # -- bucketed overlap (DDP-equivalent, reference mp_pipeline.py:92-124
    # wraps partitions in torch DDP; here the same flat buffer the
    # optimizer uses is reduced bucket-by-bucket as backward fills it) --

    def prepare_overlap(
        self,
        module: torch.nn.Module,
        group,
        divide_by: Optional[float] = None,
        bucket_mb: float = 25.0,
    ):
        """Register post-accumulate-grad hooks that all-reduce flat-buffer
        buckets as soon as backward has produced them. The engine toggles
        ``sync_enabled`` so only the LAST micro-batch triggers reduction
        (the reference relies on DDP no_sync the same way,
        train_spatial.py:1299-1308). Call ``finish_overlap`` before the
        optimizer step."""
        if group is None:
            return None
        fg = self.flat(module)
        state = {
            "fg": fg,
            "group": group,
            "divide": divide_by
            if divide_by is not None
            else float(dist.get_world_size(group=group)),
            "handles": [],
            "sync_enabled": False,
        }
        # bucket boundaries over the flat buffer, walked from the END
        # (backward fills roughly reverse parameter order)
        elems = int(bucket_mb * 1024 * 1024 / fg.buffer.element_size())
        offsets = []
        off = 0
        for p in fg.params:
            offsets.append((p, off, p.numel()))
            off += p.numel()
        buckets = []
        cur = []
        cur_elems = 0
        for item in reversed(offsets):
            cur.append(item)
            cur_elems += item[2]
            if cur_elems >= elems:
                buckets.append(cur)
                cur, cur_elems = [], 0
        if cur:
            buckets.append(cur)
        state["buckets"] = []
        for b in buckets:
            lo = min(o for _, o, _ in b)
            hi = max(o + n for _, o, n in b)
            binfo = {"lo": lo, "hi": hi, "pending": len(b), "total": len(b)}
            state["buckets"].append(binfo)
            for p, _, _ in b:
                p.register_post_accumulate_grad_hook(
                    self._mk_overlap_hook(state, binfo)
                )
        module._mpi4dl_overlap = state
        return state

    @staticmethod
    def _mk_overlap_hook(state, binfo):
        def hook(param):
            if not state["sync_enabled"]:
                return
            binfo["pending"] -= 1
            if binfo["pending"] == 0:
                binfo["pending"] = binfo["total"]
                sl = state["fg"].buffer[binfo["lo"] : binfo["hi"]]
                state["handles"].append(
                    dist.all_reduce(sl, group=state["group"], async_op=True)
                )

        return hook

    def finish_overlap(self, module: torch.nn.Module):
        state = getattr(module, "_mpi4dl_overlap", None)
        if state is None:
            return
        for h in state["handles"]:
            h.wait()
        state["handles"] = []
        state["sync_enabled"] = False
        state["fg"].rescale_(state["divide"])

    def apply_allreduce_master(self, module1, module2):
        """GEMS: average the two mirrored engines' grads over the 2-rank pair
        group, in a rank-ordered sequence so both members of a pair issue the
        two reductions in the same order (reference comm.py:460-477).
        """
        comm = self.comm
        g = comm.master_pair_groups.get(comm.rank % comm.mp_size)
        if g is None:  # degenerate mp_size==1: both engines local — average directly
            fg1, fg2 = self.flat(module1), self.flat(module2)
            if fg1.buffer.numel():
                mean = (fg1.buffer + fg2.buffer) / 2.0
                fg1.buffer.copy_(mean)
                fg2.buffer.copy_(mean)
            return
        first_half = (comm.rank % comm.mp_size) < (comm.mp_size + 1) // 2
        order = (module1, module2) if first_half else (module2, module1)
        for m in order:
            self.allreduce_grads(m, g, divide_by=2.0)
