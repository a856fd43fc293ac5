"""Sharded checkpoint/resume.

The reference has NO checkpointing (SURVEY.md §5.4 — no torch.save/load
anywhere); this module defines mpi4dl_amd's layout:

    <dir>/meta.json          — world/topology/format metadata (rank 0)
    <dir>/shard_r0007.pt     — one shard per rank:
        model:     local stage state_dict (cell-local keys)
        optimizer: optimizer state (FusedSGD flat momentum or torch sd)
        topology:  {mp_size, local_rank, split_rank, balance, spatial}
        rng:       torch CPU (+CUDA) RNG state
        model2/optimizer2: the GEMS-inverse engine's pair (if any)

``consolidate`` merges shards back into ONE full-model state_dict
(pure-LP and SP: stage keys are re-offset by the balance so cell
indices become global; replicated tile weights are taken from the first
rank of each spatial partition).
"""

from __future__ import annotations

import json
import os
import time
from typing import Optional

import torch
import torch.distributed as dist

FORMAT_VERSION = 1


def _shard_path(path: str, rank: int) -> str:
    return os.path.join(path, f"shard_r{rank:04d}.pt")


def save_checkpoint(
    path: str,
    module: torch.nn.Module,
    optimizer=None,
    comm=None,
    balance=None,
    module2: Optional[torch.nn.Module] = None,
    optimizer2=None,
    extra: Optional[dict] = None,
):
    os.makedirs(path, exist_ok=True)
    rank = dist.get_rank() if dist.is_initialized() else 0
    world = dist.get_world_size() if dist.is_initialized() else 1
    topo = {}
    if comm is not None:
        topo = {
            "mp_size": comm.mp_size,
            "dp_size": comm.dp_size,
            "local_rank": comm.local_rank,
            "split_rank": comm.split_rank,
            "spatial_parts": comm.spatial_parts,
            "spatial_size": comm.spatial_size,
            "LOCAL_DP_LP": comm.LOCAL_DP_LP,
        }
    shard = {
        "format": FORMAT_VERSION,
        "model": {k: v.cpu() for k, v in module.state_dict().items()},
        "optimizer": _opt_state(optimizer),
        "topology": topo,
        "balance": list(balance) if balance is not None else None,
        "rng": torch.get_rng_state(),
        "extra": extra or {},
    }
    if torch.cuda.is_available():
        shard["rng_cuda"] = torch.cuda.get_rng_state()
    if module2 is not None:
        shard["model2"] = {k: v.cpu() for k, v in module2.state_dict().items()}
        shard["optimizer2"] = _opt_state(optimizer2)
    torch.save(shard, _shard_path(path, rank))
    if rank == 0:
        meta = {
            "format": FORMAT_VERSION,
            "world_size": world,
            "time": time.time(),
            "topology": topo,
            "balance": list(balance) if balance is not None else None,
        }
        with open(os.path.join(path, "meta.json"), "w") as f:
            json.dump(meta, f, indent=1)
    if dist.is_initialized():
        dist.barrier()


def _opt_state(optimizer):
    if optimizer is None:
        return None
    sd = optimizer.state_dict()

    def cpu(v):
        return v.cpu() if isinstance(v, torch.Tensor) else v

    if isinstance(sd, dict):
        return {k: cpu(v) for k, v in sd.items()}
    return sd


def load_checkpoint(
    path: str,
    module: torch.nn.Module,
    optimizer=None,
    comm=None,
    module2: Optional[torch.nn.Module] = None,
    optimizer2=None,
    strict_topology: bool = True,
    restore_rng: bool = True,
) -> dict:
    rank = dist.get_rank() if dist.is_initialized() else 0
    shard = torch.load(_shard_path(path, rank), map_location="cpu", weights_only=False)
    if strict_topology and comm is not None and shard["topology"]:
        t = shard["topology"]
        assert t["mp_size"] == comm.mp_size and t["split_rank"] == comm.split_rank, (
            f"checkpoint topology {t} != runtime "
            f"(mp_size={comm.mp_size}, split_rank={comm.split_rank}); "
            "re-shard via consolidate() + a fresh partitioner"
        )
    module.load_state_dict(shard["model"])
    if optimizer is not None and shard.get("optimizer") is not None:
        optimizer.load_state_dict(shard["optimizer"])
    if module2 is not None and "model2" in shard:
        module2.load_state_dict(shard["model2"])
        if optimizer2 is not None and shard.get("optimizer2") is not None:
            optimizer2.load_state_dict(shard["optimizer2"])
    if restore_rng:
        torch.set_rng_state(shard["rng"])
        if torch.cuda.is_available() and "rng_cuda" in shard:
            torch.cuda.set_rng_state(shard["rng_cuda"])
    return shard.get("extra", {})


def consolidate(path: str) -> dict:
    """Merge all shards into one full-model state_dict (cell-global keys).

    For spatial partitions the tile weights are replicated; the first
    rank of each partition wins. GEMS shards contribute model1 only
    (model2 is the same weights by construction).
    """
    with open(os.path.join(path, "meta.json")) as f:
        meta = json.load(f)
    world = meta["world_size"]
    full: dict = {}
    seen_stages = set()
    for rank in range(world):
        shard = torch.load(_shard_path(path, rank), map_location="cpu", weights_only=False)
        topo = shard["topology"]
        balance = shard.get("balance")
        split_rank = topo.get("split_rank", rank) if topo else rank
        if split_rank in seen_stages:
            continue
        seen_stages.add(split_rank)
        # nn.Sequential slices preserve the ORIGINAL child names, so stage
        # state_dict keys are already cell-global.
        for k, v in shard["model"].items():
            full[k] = v
    return full


def load_from_consolidated(full_state: dict, module: torch.nn.Module):
    """Load a consolidated (cell-global) state_dict into a LOCAL stage of
    ANY partitioning: nn.Sequential slices keep global child names, so the
    stage's keys select its subset directly — this is the re-shard path
    (train on split=4, consolidate, resume on split=2)."""
    subset = {k: full_state[k] for k in module.state_dict().keys()}
    module.load_state_dict(subset)
