"""SpatialPlan: maps model cells -> spatial configuration for THIS rank.

The reference threads spatial config into models through a module-global
dict plus per-benchmark arithmetic (amoebanet.py:26-33,
resnet_spatial.py:272-297). Here one object owns the mapping:

* which cells are spatial (cells of partitions 0..spatial_size-1, from
  the partition balance),
* which tile this rank computes in each spatial partition,
* the tile->global-rank map, honouring GEMS inversion
  (spatial.py:913-918).

Model builders call ``plan.ctx(cell_idx)`` and get either None (build a
plain op) or kwargs for HaloConv2d / HaloPool2d / HaloExchangeLayer.
"""

from __future__ import annotations

from typing import List, Optional


class SpatialPlan:
    def __init__(
        self,
        comm,
        balance: List[int],
        slice_method: str = "square",
        grad_mode: str = "exact",
        gems_inverse: bool = False,
    ):
        self.comm = comm
        self.balance = list(balance)
        self.slice_method = slice_method
        self.grad_mode = grad_mode
        self.gems_inverse = gems_inverse
        # cell index -> partition
        self.cell_part = []
        for part, b in enumerate(self.balance):
            self.cell_part.extend([part] * b)
        self.spatial_cells = sum(self.balance[: comm.spatial_size])

    def partition_of_cell(self, cell_idx: int) -> int:
        return self.cell_part[cell_idx]

    def ctx(self, cell_idx: int) -> Optional[dict]:
        comm = self.comm
        part = self.partition_of_cell(cell_idx)
        if part >= comm.spatial_size:
            return None
        start = comm.first_local_rank_of_partition(part)
        nparts = comm.spatial_parts[part]
        # tile index of THIS rank in that partition; ranks outside the
        # partition never execute these cells for real (meta only).
        # A GEMS-inverse engine occupies the MIRRORED in-clique position
        # (reference spatial.py:913-918).
        pos = (
            comm.mp_size - 1 - comm.local_rank
            if self.gems_inverse
            else comm.local_rank
        )
        tile = pos - start
        if not (0 <= tile < nparts):
            tile = 0
        inv = self.gems_inverse

        def rank_of_tile(t, _start=start, _inv=inv):
            return comm.engine_peer(_start + t, _inv)

        return dict(
            num_spatial_parts=nparts,
            slice_method=self.slice_method,
            spatial_local_rank=tile,
            rank_of_tile=rank_of_tile,
            grad_mode=self.grad_mode,
        )

    def bn_group(self, cell_idx: int):
        """Tile process group for BN statistic sync of this cell's
        partition (None outside spatial partitions). GEMS-inverse engines
        use the mirrored groups (comm.mirror_spatial_groups)."""
        comm = self.comm
        part = self.partition_of_cell(cell_idx)
        if part >= comm.spatial_size:
            # LP partition: under local-DP, BN stats must span the DP shard
            # group to equal full-batch BN (the reference's plain BN does not)
            if comm.LOCAL_DP_LP > 1:
                return comm.local_dp_groups.get(part)
            return None
        if comm.spatial_parts[part] <= 1:
            return None
        if self.gems_inverse:
            return getattr(comm, "mirror_spatial_groups", {}).get(part)
        return comm.all_spatial_groups.get((comm.replica, part))

    def norm(self, num_features: int, cell_idx: int):
        """BatchNorm for a cell: tile-synced inside spatial partitions."""
        from .norm import TileBatchNorm2d

        g = self.bn_group(cell_idx)
        import torch.nn as nn

        if g is None and self.partition_of_cell(cell_idx) >= self.comm.spatial_size:
            return nn.BatchNorm2d(num_features)
        return TileBatchNorm2d(num_features, group=g)
