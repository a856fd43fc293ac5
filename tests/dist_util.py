"""Multi-process CPU (gloo) test harness.

Spawns ``world_size`` processes, each initialising torch.distributed
with the gloo backend over 127.0.0.1, runs ``fn(rank, world_size,
*args)``, and propagates failures. Used by every protocol-level test —
the GPU is only needed for kernels and RCCL paths (marked gpu).
"""

from __future__ import annotations

import os
import socket
import traceback

import torch.multiprocessing as mp


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world_size, port, fn, args, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        import torch.distributed as dist

        from mpi4dl_amd.comm import init_distributed

        init_distributed(backend="gloo")
        result = fn(rank, world_size, *args)
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, "ok", result))
    except BaseException:
        # BaseException: argparse's SystemExit must reach the queue too,
        # else the parent only sees a timeout
        q.put((rank, "err", traceback.format_exc()))
        raise


def run_distributed(fn, world_size: int, args: tuple = (), timeout: float = 180.0):
    """Run fn(rank, world_size, *args) on `world_size` gloo processes.

    Returns list of per-rank results (ordered by rank). Raises on any
    rank failure with its traceback.
    """
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, fn, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    errors = {}
    for _ in range(world_size):
        rank, status, payload = q.get(timeout=timeout)
        if status == "ok":
            results[rank] = payload
        else:
            errors[rank] = payload
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
    if errors:
        msgs = "\n".join(f"--- rank {r} ---\n{tb}" for r, tb in sorted(errors.items()))
        raise RuntimeError(f"distributed test failed:\n{msgs}")
    return [results[r] for r in range(world_size)]
