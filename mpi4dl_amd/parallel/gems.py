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
  replaces the pair allreduce with two flat-grad P2P legs against the
  mirror rank. Leg A (replica 1's grads — ready after its step on BOTH
  sides of the pair) is issued inside ``run_step`` between the two
  replica steps, so it rides the xGMI links while replica 2 computes;
  leg B (replica 2's grads) runs at ``allreduce_and_update``. Half the
  pair traffic is fully overlapped every iteration (the reference
  alternates odd/even roles to overlap its param leg instead,
  train_spatial_master.py:327-455).
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
        fp16_allreduce: bool = False,
        **engine_kw,
    ):
        self.comm = comm
        self.batch_size = batch_size
        self.replications = replications
        self.enable_comm_opt = enable_comm_opt
        self._fp16_allreduce = fp16_allreduce
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
        self.reducer = GradReducer(comm, fp16_allreduce=fp16_allreduce)
        self._swap = (
            _MasterOptSwap(
                comm, self.reducer, self.train_model1.models, self.train_model2.models
            )
            if enable_comm_opt
            else None
        )

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
        inputs/labels hold 2*replications*batch_size samples. With
        MASTER-OPT, replica 1's flat-grad exchange with the mirror rank
        starts here (after its last step) and overlaps replica 2's
        compute."""
        B = self.batch_size
        loss = 0.0
        corr = seen = 0
        for rep in range(self.replications):
            o = 2 * rep * B
            l1, c1, s1 = self.train_model1.run_step(
                _slc(inputs, o, B), _slc(labels, o, B)
            )
            if self._swap is not None and rep == self.replications - 1:
                self._swap.start_leg_a()
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
        if self._swap is not None:
            self._swap.finish()
        else:
            self.reducer.apply_allreduce_master(
                self.train_model1.models, self.train_model2.models
            )
        # outer DP / spatial reductions compose on top
        self.reducer.apply_allreduce(self.train_model1.models)
        self.reducer.apply_allreduce(self.train_model2.models)
        self.train_model1.update()
        self.train_model2.update()


class _MasterOptSwap:
    """MASTER-OPT flat-grad exchange with the mirror rank, split into two
    overlappable legs (reference send_recv_grads,
    train_spatial_master.py:296-325, but restructured for overlap):

    * ``start_leg_a()`` — called between the two replica steps. Both
      sides of the pair have finished their replica-1 step, so each
      sends its model1 flat grads (stage r / stage mirror) and posts the
      landing buffer for the peer's. On RCCL this batch rides the xGMI
      link while BOTH ranks compute replica 2; on gloo it completes
      inline (same semantics, no overlap — the CPU parity oracle).
    * ``finish(...)`` — called at allreduce_and_update: waits leg A,
      exchanges the model2 grads (leg B), and averages:
      my fg1 (stage r, rep1) with mirror's fg2 (stage r, rep2), and
      my fg2 (stage mirror, rep2) with mirror's fg1 (stage mirror, rep1).

    Issue-order audit (RCCL has no tags): every rank starts leg A at the
    same schedule point and leg B at the same schedule point, and the
    pipeline traffic of replica 2's step keeps its own per-pair relative
    order, so send/recv pairing is deterministic even when the mirror is
    also a pipeline neighbour (mp_size=2).
    """

    def __init__(self, comm, reducer, models1, models2):
        self.comm = comm
        self.reducer = reducer
        self.models1 = models1
        self.models2 = models2
        r = comm.rank % comm.mp_size
        self.mirror = comm.mp_size - 1 - r
        self.local = self.mirror == r
        self.peer = None if self.local else comm.global_rank(self.mirror)
        self._leg_a = None
        self._recv_for_fg2 = None  # mirror's fg1 == my fg2's stage

    def start_leg_a(self):
        fg1 = self.reducer.flat(self.models1)
        fg2 = self.reducer.flat(self.models2)
        if self.local or fg1.buffer.numel() == 0:
            return
        self._recv_for_fg2 = torch.empty_like(fg2.buffer)
        self._leg_a = p2p.exchange(
            [(fg1.buffer, self.peer, 7100)],
            [(self._recv_for_fg2, self.peer, 7100)],
        )

    def finish(self):
        fg1 = self.reducer.flat(self.models1)
        fg2 = self.reducer.flat(self.models2)
        if self.local:
            if fg1.buffer.numel():
                mean = (fg1.buffer + fg2.buffer) / 2
                fg1.buffer.copy_(mean)
                fg2.buffer.copy_(mean)
            return
        if self._leg_a is None:  # run_step never started it (e.g. eval)
            self.start_leg_a()
        if self._leg_a is None:  # empty partition: nothing to exchange
            return
        self._leg_a.wait()
        self._leg_a = None
        recv_for_fg1 = torch.empty_like(fg1.buffer)
        p2p.exchange(
            [(fg2.buffer, self.peer, 7101)],
            [(recv_for_fg1, self.peer, 7101)],
        ).wait()
        fg1.buffer.add_(recv_for_fg1).mul_(0.5)
        fg2.buffer.add_(self._recv_for_fg2).mul_(0.5)
        self._recv_for_fg2 = None


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
        fp16_allreduce: bool = False,
        **engine_kw,
    ):
        verify_spatial_master_config(comm)
        self.comm = comm
        self.batch_size = batch_size
        self.replications = replications
        self.enable_comm_opt = enable_comm_opt
        self._fp16_allreduce = fp16_allreduce
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
        self.reducer = GradReducer(comm, fp16_allreduce=fp16_allreduce)
        self._swap = (
            _MasterOptSwap(
                comm, self.reducer, self.train_model1.models, self.train_model2.models
            )
            if enable_comm_opt
            else None
        )

    def _reduce_tiles_model1(self):
        """Engine1's spatial tile sum — must precede the pair exchange so
        the mirror receives tile-complete stage grads."""
        if self.train_model1.is_tile_rank:
            self.reducer.allreduce_grads(
                self.train_model1.models,
                self.comm.spatial_allreduce_groups.get(self.train_model1.split_rank),
                divide_by=1.0,
            )

    def run_step(self, inputs, labels):
        """With MASTER-OPT: after replica 1's last step, its tile sum and
        the leg-A mirror exchange are issued here, overlapping replica
        2's compute (see _MasterOptSwap)."""
        B = self.batch_size
        loss = 0.0
        corr = seen = 0
        for rep in range(self.replications):
            o = 2 * rep * B
            l1, c1, s1 = self.train_model1.run_step(
                _slc(inputs, o, B), _slc(labels, o, B)
            )
            if self._swap is not None and rep == self.replications - 1:
                self._reduce_tiles_model1()
                self._swap.start_leg_a()
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
        # engine1 spatial groups (already reduced mid-step under MASTER-OPT)
        if self._swap is None:
            self._reduce_tiles_model1()
        # engine2: mirrored tile group of its split_rank
        if self.train_model2.is_tile_rank:
            g = getattr(comm, "mirror_spatial_groups", {}).get(
                self.train_model2.split_rank
            )
            red.allreduce_grads(self.train_model2.models, g, divide_by=1.0)
        if self._swap is not None:
            # MASTER-OPT: wait the overlapped leg A, run leg B, average
            # (reference run_step_allreduce, train_spatial_master.py:327)
            self._swap.finish()
        else:
            red.apply_allreduce_master(
                self.train_model1.models, self.train_model2.models
            )
        red.allreduce_grads(self.train_model1.models, comm.outer_dp_group)
        red.allreduce_grads(self.train_model2.models, comm.outer_dp_group)
        self.train_model1.update()
        self.train_model2.update()
