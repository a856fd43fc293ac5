"""Small shared helpers (reference parity: src/torchgems/utils.py).

Also hosts the phase-scoped tracing utility the reference lacks
(SURVEY.md §5.1): lightweight wall/GPU timers with optional roctx ranges
so rocprofv3 traces show framework phases (halo / conv / seam /
allreduce).
"""

from __future__ import annotations

import contextlib
import os
import time
from collections import defaultdict

import torch


def is_power_two(n: int) -> bool:
    """True iff n is a positive power of two (reference utils.py:20)."""
    return n > 0 and (n & (n - 1)) == 0


def get_depth(version: int, n: int) -> int:
    """ResNet depth formula: v1 = 6n+2, v2 = 9n+2 (reference utils.py:26)."""
    if version == 1:
        return n * 6 + 2
    elif version == 2:
        return n * 9 + 2
    raise ValueError(f"unknown resnet version {version}")


def env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v not in (None, "") else default


# ---------------------------------------------------------------------------
# Tracing / phase timing
# ---------------------------------------------------------------------------

_NVTX_OK = None


def _nvtx_available() -> bool:
    global _NVTX_OK
    if _NVTX_OK is None:
        _NVTX_OK = torch.cuda.is_available()
    return _NVTX_OK


class PhaseTimer:
    """Accumulates wall-clock time per named phase.

    On GPU, optionally emits roctx ranges (torch.cuda.nvtx maps to roctx on
    ROCm) so rocprofv3 --sys-trace can attribute kernels to phases. Cheap
    enough to leave on: one perf_counter pair per phase entry.
    """

    def __init__(self, use_nvtx: bool | None = None):
        self.totals: dict[str, float] = defaultdict(float)
        self.counts: dict[str, int] = defaultdict(int)
        if use_nvtx is None:
            use_nvtx = os.environ.get("MPI4DL_TRACE", "0") == "1"
        self.use_nvtx = use_nvtx and _nvtx_available()

    @contextlib.contextmanager
    def phase(self, name: str):
        if self.use_nvtx:
            torch.cuda.nvtx.range_push(name)
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self.totals[name] += time.perf_counter() - t0
            self.counts[name] += 1
            if self.use_nvtx:
                torch.cuda.nvtx.range_pop()

    def report(self) -> str:
        lines = []
        for k in sorted(self.totals, key=lambda k: -self.totals[k]):
            lines.append(
                f"{k:<24s} {self.totals[k] * 1e3:10.2f} ms  ({self.counts[k]} calls)"
            )
        return "\n".join(lines)

    def reset(self):
        self.totals.clear()
        self.counts.clear()


GLOBAL_TIMER = PhaseTimer()
