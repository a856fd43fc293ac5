"""SP(+LP+PP) training engine: spatial tiles on the leading partition(s),
pipeline stages after, with three seam topologies.

Reference parity: torchgems ``train_model_spatial``
(src/torchgems/train_spatial.py:293-1458):

* joint seam — the first LP rank receives every tile of the last spatial
  partition and concatenates the grid (reference merge_inputs_joint_cat
  :1083-1188);
* skewed seam — spatial partition k feeds spatial partition k+1 with
  fewer tiles; each coarser tile receives and merges the finer tiles it
  covers (:643-688, 1190-1254);
* local-DP seam — with LOCAL_DP_LP>1 every tile rank batch-splits its
  output across the first LP partition's DP ranks; each LP rank merges
  its batch shard of the full grid (:809-1028).

Backward mirrors every seam (tile grads scattered back, :1348-1458).

Design deviation: one generic "edge list" implementation covers all
three seams (SURVEY.md §7 notes the reference hand-writes six methods);
merges are autograd-traced ``torch.cat`` so the backward scatter falls
out of leaf gradients instead of hand-written slicing.
"""

from __future__ import annotations

import math
from typing import List, Optional

import torch

from .. import p2p
from ..comm import Communicator
from ..ops.halo import TileLayout
from ..utils import is_power_two
from .pipeline import train_model, _as_list


def verify_spatial_config(slice_method: str, image_size: int, num_spatial_parts_list):
    """Power-of-two constraints (reference train_spatial.py:33-58)."""
    assert is_power_two(image_size), f"image_size {image_size} must be a power of two"
    for p in num_spatial_parts_list:
        assert is_power_two(p), f"num_spatial_parts {p} must be a power of two"
        if slice_method == "square":
            r = int(math.isqrt(p))
            assert r * r == p, f"square slicing needs square part count, got {p}"
    # coarsening must divide evenly
    for a, b in zip(num_spatial_parts_list, num_spatial_parts_list[1:]):
        assert a % b == 0, f"spatial parts must shrink by integer factor: {a}->{b}"


def scale_shape_for_tile(shape, layout: TileLayout):
    """Full-image activation shape -> per-tile local shape."""
    n, c, h, w = shape
    assert h % layout.rows == 0 and w % layout.cols == 0, (
        f"activation {h}x{w} not divisible by tile grid {layout.rows}x{layout.cols}"
    )
    return (n, c, h // layout.rows, w // layout.cols)


def get_shapes_spatial(shape_list, slice_method, spatial_size, num_spatial_parts_list):
    """Per-stage LOCAL output shapes (reference train_spatial.py:61-238).

    For spatial stages the full-image shape (from meta inference) is
    divided by that stage's tile grid; LP stages keep the full shape.
    """
    out = []
    for i, spec in enumerate(shape_list):
        specs = _as_list(spec)
        if i < spatial_size:
            layout = TileLayout(num_spatial_parts_list[i], slice_method)
            specs = [scale_shape_for_tile(s, layout) for s in specs]
        out.append(specs if len(specs) > 1 else specs[0])
    return out


class train_model_spatial(train_model):
    """SP+LP engine for one rank. ``model_gen.shape_list`` must hold the
    FULL-image per-stage shapes (meta inference); this class derives all
    local/seam shapes from the tile layouts."""

    def __init__(
        self,
        model_gen,
        local_rank: int,
        batch_size: int,
        parts: int,
        comm: Communicator,
        slice_method: str = "square",
        grad_mode: str = "exact",
        **kw,
    ):
        self.slice_method = slice_method
        self.grad_mode = grad_mode
        assert comm.ENABLE_SPATIAL and comm.spatial_size >= 1
        self.spatial_size = comm.spatial_size
        self.layouts = [
            TileLayout(p, slice_method) for p in comm.spatial_parts
        ]
        # train_model.__init__ calls _init_peers/_init_buffers which we override
        super().__init__(model_gen, local_rank, batch_size, parts, comm, **kw)

    # ------------------------------------------------------------------
    # roles & peers
    # ------------------------------------------------------------------

    def _init_peers(self):
        comm = self.comm
        sr = self.split_rank
        self.first_stage = sr == 0
        self.last_stage = sr == self.split_size - 1
        self.is_tile_rank = sr < self.spatial_size
        self.tile_idx = (
            self.local_rank - comm.first_local_rank_of_partition(sr)
            if self.is_tile_rank
            else 0
        )
        self.layout = self.layouts[sr] if self.is_tile_rank else None
        L = comm.LOCAL_DP_LP
        self.my_dp = (
            self.local_rank - comm.first_local_rank_of_partition(sr)
            if (not self.is_tile_rank and L > 1)
            else 0
        )

        def peer(pos):
            return comm.engine_peer(pos, self.GEMS_INVERSE)

        # ---- forward-recv / backward-send edges (who feeds me) ----------
        # list of (global_rank, kind) where kind tags the merge slot
        self.in_edges: List[int] = []
        self.in_layout: Optional[TileLayout] = None  # grid of incoming tiles
        self.in_grid: List[List[int]] = []  # row-major [rows][cols] edge idx
        if not self.first_stage:
            prev = sr - 1
            prev_start = comm.first_local_rank_of_partition(prev)
            if self.is_tile_rank:
                # skewed seam: finer grid -> my coarser tile
                fine, coarse = self.layouts[prev], self.layout
                rr, cc = fine.rows // coarse.rows, fine.cols // coarse.cols
                r0, c0 = coarse.pos(self.tile_idx)
                self.in_layout = fine
                self.in_sub = (rr, cc)
                for dr in range(rr):
                    row = []
                    for dc in range(cc):
                        t = fine.tile_at(r0 * rr + dr, c0 * cc + dc)
                        row.append(len(self.in_edges))
                        self.in_edges.append(peer(prev_start + t))
                    self.in_grid.append(row)
            elif prev < self.spatial_size:
                # joint (or local-DP) seam: all tiles of last spatial stage
                fine = self.layouts[prev]
                self.in_layout = fine
                self.in_sub = (fine.rows, fine.cols)
                for r in range(fine.rows):
                    row = []
                    for c in range(fine.cols):
                        t = fine.tile_at(r, c)
                        row.append(len(self.in_edges))
                        self.in_edges.append(peer(prev_start + t))
                    self.in_grid.append(row)
            else:
                # plain pipeline edge (same local-DP position)
                my_dp = (
                    self.local_rank - comm.first_local_rank_of_partition(sr)
                ) if L > 1 else 0
                self.in_edges = [peer(prev_start + my_dp)]
        self.prev_rank = self.in_edges[0] if self.in_edges else None

        # ---- forward-send / backward-recv edges (whom I feed) -----------
        self.out_edges: List[int] = []
        self.out_mode = "none"
        if not self.last_stage:
            nxt = sr + 1
            nxt_start = comm.first_local_rank_of_partition(nxt)
            if nxt < self.spatial_size:
                # skewed: my tile feeds exactly one coarser tile
                fine, coarse = self.layout, self.layouts[nxt]
                rr, cc = fine.rows // coarse.rows, fine.cols // coarse.cols
                r, c = fine.pos(self.tile_idx)
                t = coarse.tile_at(r // rr, c // cc)
                self.out_edges = [peer(nxt_start + t)]
                self.out_mode = "skewed"
            elif self.is_tile_rank:
                # joint seam: send my tile to first-LP rank(s)
                self.out_edges = [peer(nxt_start + d) for d in range(L)]
                self.out_mode = "joint" if L == 1 else "joint_dp"
            else:
                my_dp = (
                    self.local_rank - comm.first_local_rank_of_partition(sr)
                ) if L > 1 else 0
                self.out_edges = [peer(nxt_start + my_dp)]
                self.out_mode = "pipeline"
        self.next_rank = self.out_edges[0] if self.out_edges else None

    # ------------------------------------------------------------------
    # buffers
    # ------------------------------------------------------------------

    def _local_mb(self) -> int:
        """My stage's micro-batch size (LP stages under local-DP see 1/L)."""
        L = self.comm.LOCAL_DP_LP
        if not self.is_tile_rank and L > 1:
            assert self.mb % L == 0, (
                f"micro-batch {self.mb} not divisible by LOCAL_DP_LP {L}: "
                "raise batch_size or lower parts/local-DP"
            )
            return self.mb // L
        return self.mb

    def _init_buffers(self):
        self.input_buffers = None
        self.grad_buffers = None
        comm = self.comm
        L = comm.LOCAL_DP_LP
        if not self.first_stage:
            prev = self.split_rank - 1
            spec = _as_list(self.shape_list[prev])
            if self.in_layout is not None:
                # seam tags stride 4 per edge (100 + i*4 + j below and the
                # 200+ grad mirrors): wider tuples would collide on gloo
                assert len(spec) <= 4, (
                    f"stage boundary carries a {len(spec)}-tensor tuple; the "
                    "seam tag stride supports at most 4 (widen the stride in "
                    "receive_input/send_input_grad to raise this)"
                )
                # per-edge x per-tensor buffers (tuple activations supported:
                # each tile sends every tensor of the tuple)
                mb = self._local_mb()
                tile_shapes = [
                    (mb,) + tuple(scale_shape_for_tile(sp, self.in_layout)[1:])
                    for sp in spec
                ]
                self.input_buffers = [
                    [
                        [
                            torch.zeros(ts, device=self.device, dtype=self.act_dtype)
                            for ts in tile_shapes
                        ]
                        for _ in self.in_edges
                    ]
                    for _ in range(self.parts)
                ]
            else:
                shapes = [
                    (self._local_mb(),) + tuple(s[1:]) for s in spec
                ]
                self.input_buffers = [
                    [
                        torch.zeros(s, device=self.device, dtype=self.act_dtype)
                        for s in shapes
                    ]
                    for _ in range(self.parts)
                ]
        if not self.last_stage:
            spec = _as_list(self.shape_list[self.split_rank])
            if self.is_tile_rank:
                my_shapes = [
                    tuple(scale_shape_for_tile(sp, self.layout)) for sp in spec
                ]
                if self.out_mode == "joint_dp":
                    mb = self.mb // L
                    # per-DP-peer x per-tensor
                    self.grad_buffers = [
                        [
                            [
                                torch.zeros(
                                    (mb,) + tuple(ms[1:]),
                                    device=self.device,
                                    dtype=self.act_dtype,
                                )
                                for ms in my_shapes
                            ]
                            for _ in range(L)
                        ]
                        for _ in range(self.parts)
                    ]
                    return
                shapes = [(self.mb,) + tuple(ms[1:]) for ms in my_shapes]
            else:
                shapes = [(self._local_mb(),) + tuple(s[1:]) for s in spec]
            self.grad_buffers = [
                [
                    torch.zeros(s, device=self.device, dtype=self.act_dtype)
                    for s in shapes
                ]
                for _ in range(self.parts)
            ]

    # ------------------------------------------------------------------
    # forward seams
    # ------------------------------------------------------------------

    def receive_input(self, part: int):
        bufs = self.input_buffers[part]
        if self.in_layout is None:
            p2p.recv_tensors(bufs, self.prev_rank, tag_base=1000 + part * 16)
            leaves = [self._leaf(b) for b in bufs]
            return leaves[0] if len(leaves) == 1 else tuple(leaves)
        # gather tiles (joint / skewed / joint_dp receiving side);
        # bufs[i][j] = tensor j of the tuple from edge i
        recvs = [
            (b, peer, 100 + i * 4 + j)
            for i, peer in enumerate(self.in_edges)
            for j, b in enumerate(bufs[i])
        ]
        p2p.exchange([], recvs).wait()
        leaves = [[self._leaf(b) for b in eb] for eb in bufs]
        self._seam_leaves = getattr(self, "_seam_leaves", [None] * self.parts)
        self._seam_leaves[part] = leaves
        # merge grid per tuple slot: cat cols within row, then rows
        # (reference merge_inputs_joint_cat, train_spatial.py:1083)
        ntens = len(bufs[0])
        merged = []
        for j in range(ntens):
            rows = [
                torch.cat([leaves[i][j] for i in row], dim=3)
                for row in self.in_grid
            ]
            merged.append(torch.cat(rows, dim=2) if len(rows) > 1 else rows[0])
        return merged[0] if ntens == 1 else tuple(merged)

    def send_output(self, y, part: int):
        ys = list(y) if isinstance(y, tuple) else [y]
        if self.out_mode == "joint_dp":
            # batch-split across the local-DP ranks (reference :809-853)
            sends = []
            for j, t in enumerate(ys):
                chunks = t.to(self.act_dtype).chunk(len(self.out_edges), dim=0)
                for c, peer in zip(chunks, self.out_edges):
                    sends.append(
                        (c.contiguous(), peer, 100 + self._joint_dp_slot() * 4 + j)
                    )
            self._pending.append(p2p.exchange(sends, []))
        elif self.out_mode in ("joint", "skewed"):
            slot = self._seam_slot()
            sends = [
                (t.to(self.act_dtype).contiguous(), self.out_edges[0], 100 + slot * 4 + j)
                for j, t in enumerate(ys)
            ]
            self._pending.append(p2p.exchange(sends, []))
        else:
            tr = p2p.isend_tensors(
                [t.to(self.act_dtype) for t in ys],
                self.out_edges[0],
                tag_base=1000 + part * 16,
            )
            self._pending.append(tr)

    def _seam_slot(self) -> int:
        """My edge index at the receiving rank's in_edges (row-major grid)."""
        comm = self.comm
        nxt = self.split_rank + 1
        if nxt < self.spatial_size:
            fine, coarse = self.layout, self.layouts[nxt]
            rr, cc = fine.rows // coarse.rows, fine.cols // coarse.cols
            r, c = fine.pos(self.tile_idx)
            return (r % rr) * cc + (c % cc)
        return self.tile_idx

    def _joint_dp_slot(self) -> int:
        return self.tile_idx

    # ------------------------------------------------------------------
    # backward seams
    # ------------------------------------------------------------------

    def receive_output_grad(self, part: int):
        bufs = self.grad_buffers[part]
        if self.out_mode in ("joint", "skewed"):
            slot = self._seam_slot()
            recvs = [
                (b, self.out_edges[0], 200 + slot * 4 + j)
                for j, b in enumerate(bufs)
            ]
            p2p.exchange([], recvs).wait()
            return bufs
        if self.out_mode == "joint_dp":
            recvs = [
                (b, peer, 200 + self._joint_dp_slot() * 4 + j)
                for d, peer in enumerate(self.out_edges)
                for j, b in enumerate(bufs[d])
            ]
            p2p.exchange([], recvs).wait()
            # each local-DP rank's loss is the mean over its mb/L shard;
            # the full-batch mean is 1/L of the shard-mean sum — rescale
            # so tile-side gradients match the serial trajectory.
            L = len(self.out_edges)
            ntens = len(bufs[0])
            return [
                torch.cat([bufs[d][j] for d in range(L)], dim=0).div_(L)
                for j in range(ntens)
            ]
        p2p.recv_tensors(bufs, self.next_rank, tag_base=3000 + part * 16)
        return bufs

    def send_input_grad(self, part: int):
        if self.in_layout is None:
            return super().send_input_grad(part)
        leaves = self._seam_leaves[part]
        sends = []
        for i, (edge_leaves, peer) in enumerate(zip(leaves, self.in_edges)):
            for j, leaf in enumerate(edge_leaves):
                g = (
                    leaf.grad if leaf.grad is not None else torch.zeros_like(leaf)
                ).to(self.act_dtype)
                sends.append((g.contiguous(), peer, 200 + i * 4 + j))
        self._pending.append(p2p.exchange(sends, []))
        self._seam_leaves[part] = None

    # ------------------------------------------------------------------
    # step
    # ------------------------------------------------------------------

    def _slice_io(self, inputs, labels):
        """First-stage tile ranks slice their tile from the full input
        (reference split_input, train_spatial.py:241); local-DP last
        stages shard the labels."""
        if inputs is not None and self.first_stage and self.is_tile_rank:
            inputs = self.layout.slice_input(inputs, self.tile_idx).contiguous()
        L = self.comm.LOCAL_DP_LP
        if (
            labels is not None
            and self.last_stage
            and not self.is_tile_rank
            and L > 1
        ):
            # my batch shard of each micro-batch's labels (local-DP seam)
            shards = [
                py.chunk(L, dim=0)[self.my_dp] for py in labels.chunk(self.parts, 0)
            ]
            labels = torch.cat(shards, dim=0)
        return inputs, labels

    def run_step(self, inputs, labels):
        inputs, labels = self._slice_io(inputs, labels)
        return super().run_step(inputs, labels)

    @torch.no_grad()
    def run_eval(self, inputs, labels):
        inputs, labels = self._slice_io(inputs, labels)
        return super().run_eval(inputs, labels)
