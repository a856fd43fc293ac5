"""The torchgems-style benchmark entry points (C22-C25) end-to-end on
gloo: every runner mode — lp / sp / gems / gems_sp — with both model
families, at tiny config. Reference parity: the per-benchmark main()
bodies under /root/reference/benchmarks/* (e.g.
benchmark_resnet_gems_master.py:97-309); here one shared runner serves
all entries, so driving run_training per mode covers each entry's
code path."""

import os
import sys

from dist_util import run_distributed


def _run_mode(rank, world, mode, model, extra):
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from runner import run_training

    from mpi4dl_amd.parser import get_parser

    argv = [
        "--model", model, "--batch-size", "4", "--parts", "2",
        "--image-size", "32", "--num-layers", "9", "--num-filters", "4",
        "--num-classes", "10", "--num-epochs", "1", "--num-steps", "2",
        "--backend", "gloo",
    ] + list(extra)
    args = get_parser().parse_args(argv)
    times = run_training(args, mode)
    return len(times)


def test_entry_lp_resnet():
    # 2 LP stages (benchmark_resnet_lp.py equivalent)
    got = run_distributed(
        _run_mode, 2,
        ("lp", "resnet", ("--split-size", "2", "--enable-evaluation")))
    assert got[0] == 2


def test_entry_sp_amoebanet():
    # 2 spatial tiles + 1 LP stage (benchmark_amoebanet_sp.py equivalent)
    got = run_distributed(
        _run_mode, 3,
        ("sp", "amoebanet",
         ("--split-size", "2", "--num-spatial-parts", "2",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--num-layers", "6", "--num-filters", "8",
          # 128²: the deepest spatial cell's tile must fit the 1×7 halo
          "--image-size", "128")),
        timeout=300,
    )
    assert got[0] == 2


def test_entry_gems_resnet():
    # two mirrored LP engines on 2 ranks (benchmark_resnet_gems_master.py)
    got = run_distributed(
        _run_mode, 2,
        ("gems", "resnet", ("--split-size", "2", "--enable-evaluation")))
    assert got[0] == 2


def test_entry_gems_sp_resnet():
    # GEMS on top of 2 spatial tiles + 2 LP stages: mp = 3 + 2 - 1 = 4,
    # satisfying the mp >= 2*tiles mirror-disjointness requirement
    got = run_distributed(
        _run_mode, 4,
        ("gems_sp", "resnet",
         ("--split-size", "3", "--num-spatial-parts", "2",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--enable-evaluation")),
        timeout=300,
    )
    assert got[0] == 2


def test_entry_gems_sp_comm_opt():
    # MASTER-OPT overlapped gradient swap path
    got = run_distributed(
        _run_mode, 4,
        ("gems_sp", "resnet",
         ("--split-size", "3", "--num-spatial-parts", "2",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--enable-master-comm-opt")),
        timeout=300,
    )
    assert got[0] == 2


def test_entry_sp_d2_resnet():
    # D2 fused-halo model through the sp entry (--halo-D2)
    got = run_distributed(
        _run_mode, 3,
        ("sp", "resnet",
         ("--split-size", "2", "--num-spatial-parts", "2",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--halo-d2", "--image-size", "64")),
        timeout=300,
    )
    assert got[0] == 2


def test_entry_lp_adamw():
    # --optimizer / --weight-decay are honoured (adamw path)
    got = run_distributed(
        _run_mode, 2,
        ("lp", "resnet",
         ("--split-size", "2", "--optimizer", "adamw",
          "--weight-decay", "0.01")))
    assert got[0] == 2


def test_entry_sp_resnet101():
    # BASELINE config 4 shape: ResNet-101 cells with SP+PP (tiny size)
    got = run_distributed(
        _run_mode, 3,
        ("sp", "resnet101",
         ("--split-size", "2", "--num-spatial-parts", "2",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--image-size", "64", "--batch-size", "2", "--parts", "1")),
        timeout=300,
    )
    assert got[0] == 2


def test_entry_gems_times2():
    # --times 2: two replica pairs per step (4x batch consumed)
    got = run_distributed(
        _run_mode, 2,
        ("gems", "resnet", ("--split-size", "2", "--times", "2")))
    assert got[0] == 2
