"""C26 halo micro-benchmark entry points end-to-end on gloo: all three
modes (exchange / with-compute / conv) with their built-in exact
validation enabled."""

import io
import os
import sys

from dist_util import run_distributed


def _bench_body(rank, world, mode, slice_method):
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks", "communication", "halo"))
    import halo_bench

    argv = [
        "halo_bench.py", "--mode", mode, "--image-size", "64",
        "--halo-len", "2", "--channels", "3", "--out-channels", "8",
        "--iterations", "3", "--warmup", "1",
        "--slice-method", slice_method,
    ]
    old_argv, old_out = sys.argv, sys.stdout
    sys.argv = argv
    sys.stdout = cap = io.StringIO()
    try:
        halo_bench.main()
    finally:
        sys.argv, sys.stdout = old_argv, old_out
    return cap.getvalue()


def test_halo_bench_exchange():
    out = run_distributed(_bench_body, 4, ("exchange", "vertical"))
    # each rank prints its validation verdict; assert + no FAILED
    assert all("Validation passed" in o for o in out), out


def test_halo_bench_compute():
    out = run_distributed(_bench_body, 2, ("compute", "horizontal"))
    assert all("Validation passed" in o for o in out), out


def test_halo_bench_conv_square():
    out = run_distributed(_bench_body, 4, ("conv", "square"))
    assert all("Validation passed" in o for o in out), out
