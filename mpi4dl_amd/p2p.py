"""Point-to-point transport over RCCL (or gloo for CPU tests).

RCCL has no message tags; correctness comes from deterministic issue
order between each pair of ranks (one process per GPU, SPMD schedule).
On the nccl backend all sends/recvs of one logical transfer are issued
as ONE ``batch_isend_irecv`` group → a single grouped ncclSend/ncclRecv
launch over xGMI. On gloo we fall back to tagged isend/irecv, which
keeps multi-tensor transfers unambiguous in CPU tests.

Replaces the reference's tag discipline (spatial.py:170-175) and its
blocking ``req.wait()`` + ``torch.cuda.synchronize()`` fences
(spatial.py:377-383): RCCL ops are stream-ordered, so no host fences are
needed between comm and compute on the same stream.
"""

from __future__ import annotations

import os
import time
from typing import List, Sequence

import torch
import torch.distributed as dist

# Failure detection (the reference hangs forever on any
# rank failure). MPI4DL_WAIT_TIMEOUT=<seconds> arms a watchdog on every
# P2P wait; on expiry we raise with the pending peer set instead of
# deadlocking the whole job. 0 (default) = off (zero overhead).
_WAIT_TIMEOUT = float(os.environ.get("MPI4DL_WAIT_TIMEOUT", "0") or 0)


def _is_nccl() -> bool:
    return dist.get_backend() == "nccl"


class Transfer:
    """Handle for an in-flight batched transfer."""

    def __init__(self, reqs, peers=None):
        self.reqs = reqs
        self.peers = peers or []

    def wait(self):
        if _WAIT_TIMEOUT > 0:
            deadline = time.monotonic() + _WAIT_TIMEOUT
            for r in self.reqs:
                while not r.is_completed():
                    if time.monotonic() > deadline:
                        raise RuntimeError(
                            f"P2P wait exceeded {_WAIT_TIMEOUT}s "
                            f"(peers={sorted(set(self.peers))}): a peer rank "
                            "is stuck or dead (MPI4DL_WAIT_TIMEOUT watchdog)"
                        )
                    time.sleep(0.005)
            self.reqs = []
            return
        for r in self.reqs:
            r.wait()
        self.reqs = []


def isend_tensors(tensors: Sequence[torch.Tensor], peer: int, tag_base: int = 0) -> Transfer:
    if _is_nccl():
        ops = [dist.P2POp(dist.isend, t, peer) for t in tensors]
        return Transfer(dist.batch_isend_irecv(ops), [peer]) if ops else Transfer([])
    reqs = [dist.isend(t, peer, tag=tag_base + i) for i, t in enumerate(tensors)]
    return Transfer(reqs, [peer])


def irecv_tensors(buffers: Sequence[torch.Tensor], peer: int, tag_base: int = 0) -> Transfer:
    if _is_nccl():
        ops = [dist.P2POp(dist.irecv, t, peer) for t in buffers]
        return Transfer(dist.batch_isend_irecv(ops), [peer]) if ops else Transfer([])
    reqs = [dist.irecv(t, peer, tag=tag_base + i) for i, t in enumerate(buffers)]
    return Transfer(reqs, [peer])


def exchange(
    send_list: List[tuple],  # (tensor, peer, tag)
    recv_list: List[tuple],  # (buffer, peer, tag)
) -> Transfer:
    """Issue a mixed batch of sends+recvs as one grouped call (halo pattern).

    With RCCL this is one ncclGroupStart/End — all 2–9 neighbour messages
    of a halo exchange ride their xGMI links concurrently; tags are
    ignored (per-pair ordering disambiguates). On gloo the caller-chosen
    tags pair up sender/receiver sides (tags must agree on both ends —
    e.g. the *direction as seen by the receiver* for halos).
    """
    if _is_nccl():
        # NCCL P2P between the same pair must be issued in a globally
        # consistent order; sort by (peer, tag) on both sides.
        ops = [
            dist.P2POp(dist.isend, t, p)
            for t, p, _ in sorted(send_list, key=lambda x: (x[1], x[2]))
        ]
        ops += [
            dist.P2POp(dist.irecv, b, p)
            for b, p, _ in sorted(recv_list, key=lambda x: (x[1], x[2]))
        ]
        peers = [p for _, p, _ in send_list] + [p for _, p, _ in recv_list]
        return Transfer(dist.batch_isend_irecv(ops), peers) if ops else Transfer([])
    reqs = []
    for t, p, tag in send_list:
        reqs.append(dist.isend(t, p, tag=tag))
    for b, p, tag in recv_list:
        reqs.append(dist.irecv(b, p, tag=tag))
    return Transfer(reqs, [p for _, p, _ in send_list] + [p for _, p, _ in recv_list])


def send_tensors(tensors, peer, tag_base: int = 0):
    isend_tensors(tensors, peer, tag_base).wait()


def recv_tensors(buffers, peer, tag_base: int = 0):
    irecv_tensors(buffers, peer, tag_base).wait()
