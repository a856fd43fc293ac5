"""GEMS (bidirectional model parallelism) engines.

Reference parity: torchgems ``train_model_master``
(src/torchgems/gems_master.py:23-103) and
``train_spatial_model_master`` (src/torchgems/train_spatial_master.py).

GEMS runs TWO replicas of the pipelined model on the SAME GPUs in
opposite rank order: replica 2's stage j lives on rank mp-1-j
(comm.py:77-80, mp_pipeline.py:244-248). While replica 1's forward
sweeps ranks 0->n-1, replica 2's sweeps n-1->0, so the pipeline bubbles
of one replica are filled by the other — ~2x effective batch at the
same per-GPU weight memory plus one extra stage's weights.

MI355X-native notes:
* engines share the GPU; their P2P edges use the same RCCL communicator
  with deterministic per-pair ordering (engine1's step fully precedes
  engine2's step, as in the reference's interleave, gems_master.py:72-103);
* gradient pairing runs over the 2-rank {j, mp-1-j} groups in the
  rank-ordered sequence that avoids deadlock (comm.py:460-477) — see
  GradReducer.apply_allreduce_master;
* MASTER-OPT (reference run_step_allreduce, train_spatial_master.py:327)
  is implemented as an OVERLAPPED flat-grad swap: replica 1's grad
  exchange with the mirror rank rides the network while replica 2 is
  still computing.
"""

from __future__ import annotations

import torch

from .. import p2p
from ..comm import Communicator, GradReducer
from .pipeline import train_model
from .spatial import train_model_spatial


def verify_spatial_master_config(comm: Communicator):
    """GEMS+SP needs mp_size >= 2 * spatial tile count so the two
    replicas' tile ranks don't collide (reference
    train_spatial_master.py:33-84)."""
    total_tiles = comm.total_spatial_ranks
    assert comm.mp_size >= 2 * total_tiles, (
        f"GEMS+SP requires mp_size ({comm.mp_size}) >= 2*spatial tiles "
        f"({total_tiles}): replica 2's mirrored tile ranks must not overlap "
        "replica 1's"
    )


class train_model_master:
    """Two mirrored LP+PP engines on this rank (reference gems_master.py).

    ``model_gen1``/``model_gen2`` must be ready_model()'d for positions
    ``comm.local_rank`` and ``mp_size-1-comm.local_rank`` respectively.
    """

    def __init__(
        self,
        model_gen1,
        model_gen2,
        batch_size: int,
        parts: int,
        comm: Communicator,
        replications: int = 1,
        enable_comm_opt: bool = False,
        **engine_kw,
    ):
        self.comm = comm
        self.batch_size = batch_size
        self.replications = replications
        self.enable_comm_opt = enable_comm_opt
        r = comm.rank % comm.mp_size
        self.train_model1 = train_model(
            model_gen1, r, batch_size, parts, comm, GEMS_INVERSE=False, **engine_kw
        )
        self.train_model2 = train_model(
            model_gen2,
            comm.mp_size - 1 - r,
            batch_size,
            parts,
            comm,
            GEMS_INVERSE=True,
            **engine_kw,
        )
        self.reducer = GradReducer(comm)

    # -- weight sync at init (reference comm.py:382-400) -------------------

    def sync_models(self):
        """Make replica 2's weights equal replica 1's via the mirror pair
        (models built from one seed are already equal; this guards
        against divergent init)."""
        comm = self.comm
        r = comm.rank % comm.mp_size
        mirror = comm.mp_size - 1 - r
        if mirror == r:
            self.train_model2.models.load_state_dict(
                self.train_model1.models.state_dict()
            )
            return
        peer = comm.global_rank(mirror)
        params1 = [p.data for p in self.train_model1.models.parameters()]
        params2 = [p.data for p in self.train_model2.models.parameters()]
        # my model1 stage r pairs with mirror's model2 stage r: send mine,
        # receive theirs (which overwrites my model2 stage `mirror`).
        flat1 = torch.cat([p.reshape(-1) for p in params1]) if params1 else None
        if params2:
            recv = torch.empty(
                sum(p.numel() for p in params2),
                device=params2[0].device,
                dtype=params2[0].dtype,
            )
        first = r < mirror
        if first:
            if flat1 is not None:
                p2p.send_tensors([flat1], peer, tag_base=7000)
            if params2:
                p2p.recv_tensors([recv], peer, tag_base=7001)
        else:
            if params2:
                p2p.recv_tensors([recv], peer, tag_base=7000)
            if flat1 is not None:
                p2p.send_tensors([flat1], peer, tag_base=7001)
        if params2:
            off = 0
            for p in params2:
                n = p.numel()
                p.copy_(recv[off : off + n].view_as(p))
                off += n

    # -- stepping -----------------------------------------------------------

    def run_step(self, inputs, labels):
        """Interleave replica steps (reference gems_master.py:72-103):
        inputs/labels hold 2*replications*batch_size samples."""
        B = self.batch_size
        loss = 0.0
        corr = seen = 0
        for rep in range(self.replications):
            o = 2 * rep * B
            l1, c1, s1 = self.train_model1.run_step(
                _slc(inputs, o, B), _slc(labels, o, B)
            )
            l2, c2, s2 = self.train_model2.run_step(
                _slc(inputs, o + B, B), _slc(labels, o + B, B)
            )
            loss += (l1 + l2) / 2
            corr += c1 + c2
            seen += s1 + s2
        return loss / self.replications, corr, seen

    @torch.no_grad()
    def run_eval(self, inputs, labels):
        """Forward-only pass on replica 1 only — after
        allreduce_and_update both replicas hold identical weights, so
        one engine's eval is the model's eval (inputs/labels: a plain
        batch_size batch, not the 2x training layout)."""
        return self.train_model1.run_eval(inputs, labels)

    def allreduce_and_update(self):
        """Pair the two replicas' grads, then step both optimizers
        (reference apply_allreduce_master_and_update, comm.py:516-523)."""
        if self.enable_comm_opt:
            self._paired_swap()
        else:
            self.reducer.apply_allreduce_master(
                self.train_model1.models, self.train_model2.models
            )
        # outer DP / spatial reductions compose on top
        self.reducer.apply_allreduce(self.train_model1.models)
        self.reducer.apply_allreduce(self.train_model2.models)
        self.train_model1.update()
        self.train_model2.update()

    def _paired_swap(self):
        _paired_grad_swap(self.comm, self.reducer,
                          self.train_model1.models, self.train_model2.models)


def _paired_grad_swap(comm, reducer, models1, models2):
    """MASTER-OPT: flat-grad P2P swap with the mirror rank instead of
    a pair allreduce (reference send_recv_grads,
    train_spatial_master.py:296-325)."""
    r = comm.rank % comm.mp_size
    mirror = comm.mp_size - 1 - r
    fg1 = reducer.flat(models1)
    fg2 = reducer.flat(models2)
    if mirror == r:
        if fg1.buffer.numel():
            mean = (fg1.buffer + fg2.buffer) / 2
            fg1.buffer.copy_(mean)
            fg2.buffer.copy_(mean)
        return
    peer = comm.global_rank(mirror)
    # my model1 grads pair with mirror's model2 grads (same stage)
    r1 = torch.empty_like(fg1.buffer)
    r2 = torch.empty_like(fg2.buffer)
    # tags keyed by STAGE index (my model1 holds stage r, my model2
    # stage `mirror`): the pair exchanges stage min(r,mirror) first on
    # both sides, so gloo tags AND RCCL issue order both line up.
    first = r < mirror
    seq = (
        [(fg1.buffer, r1, 7100), (fg2.buffer, r2, 7101)]
        if first
        else [(fg2.buffer, r2, 7100), (fg1.buffer, r1, 7101)]
    )
    # both sides order the two swaps by stage index (min stage first),
    # so RCCL's order-based pairing matches: my model1<->mirror model2
    # then my model2<->mirror model1.
    for send_buf, recv_buf, tag in seq:
        p2p.exchange([(send_buf, peer, tag)], [(recv_buf, peer, tag)]).wait()
    fg1.buffer.add_(r1).mul_(0.5)
    fg2.buffer.add_(r2).mul_(0.5)


def _slc(t, off, n):
    return None if t is None else t[off : off + n]


class train_spatial_model_master:
    """GEMS composed with SP (reference train_spatial_master.py:87-501):
    two mirrored train_model_spatial engines + paired reductions."""

    def __init__(
        self,
        model_gen1,
        model_gen2,
        batch_size: int,
        parts: int,
        comm: Communicator,
        slice_method: str = "square",
        replications: int = 1,
        enable_comm_opt: bool = False,
        **engine_kw,
    ):
        verify_spatial_master_config(comm)
        self.comm = comm
        self.batch_size = batch_size
        self.replications = replications
        self.enable_comm_opt = enable_comm_opt
        r = comm.rank % comm.mp_size
        self.train_model1 = train_model_spatial(
            model_gen1, r, batch_size, parts, comm,
            slice_method=slice_method, GEMS_INVERSE=False, **engine_kw,
        )
        self.train_model2 = train_model_spatial(
            model_gen2,
            comm.mp_size - 1 - r,
            batch_size,
            parts,
            comm,
            slice_method=slice_method,
            GEMS_INVERSE=True,
            **engine_kw,
        )
        self.reducer = GradReducer(comm)

    def run_step(self, inputs, labels):
        B = self.batch_size
        loss = 0.0
        corr = seen = 0
        for rep in range(self.replications):
            o = 2 * rep * B
            l1, c1, s1 = self.train_model1.run_step(
                _slc(inputs, o, B), _slc(labels, o, B)
            )
            l2, c2, s2 = self.train_model2.run_step(
                _slc(inputs, o + B, B), _slc(labels, o + B, B)
            )
            loss += (l1 + l2) / 2
            corr += c1 + c2
            seen += s1 + s2
        return loss / self.replications, corr, seen

    @torch.no_grad()
    def run_eval(self, inputs, labels):
        """Forward-only pass on replica 1 (see train_model_master.run_eval)."""
        return self.train_model1.run_eval(inputs, labels)

    def allreduce_and_update(self):
        """Spatial tile reduction per engine (engine2 over mirror groups),
        then the GEMS pair reduction, then update (reference
        apply_allreduce_master_master, comm.py:479-504)."""
        comm = self.comm
        red = self.reducer
        # engine1 spatial groups
        if self.train_model1.is_tile_rank:
            red.allreduce_grads(
                self.train_model1.models,
                comm.spatial_allreduce_groups.get(self.train_model1.split_rank),
                divide_by=1.0,
            )
        # engine2: mirrored tile group of its split_rank
        if self.train_model2.is_tile_rank:
            g = getattr(comm, "mirror_spatial_groups", {}).get(
                self.train_model2.split_rank
            )
            red.allreduce_grads(self.train_model2.models, g, divide_by=1.0)
        if self.enable_comm_opt:
            # MASTER-OPT: the mirror-pair exchange as a flat P2P swap
            # (reference run_step_allreduce, train_spatial_master.py:327)
            _paired_grad_swap(comm, red, self.train_model1.models,
                              self.train_model2.models)
        else:
            red.apply_allreduce_master(
                self.train_model1.models, self.train_model2.models
            )
        red.allreduce_grads(self.train_model1.models, comm.outer_dp_group)
        red.allreduce_grads(self.train_model2.models, comm.outer_dp_group)
        self.train_model1.update()
        self.train_model2.update()
