"""BASELINE.json named configs, exercised at their CPU-checkable shape
(the GPU-scale variants run through the same code paths at round end):

1. ResNet-18 layer parallelism, 224x224, world_size=2 on CPU/gloo
3. AmoebaNet-D GEMS bidirectional pipeline, 8 stages (512^2 on GPUs;
   tiny filters here — the 8-stage GEMS schedule is what's validated)
"""

import os
import sys

from dist_util import run_distributed


def _run_mode(rank, world, mode, extra):
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from runner import run_training

    from mpi4dl_amd.parser import get_parser

    args = get_parser().parse_args([
        "--num-epochs", "1", "--num-steps", "2", "--backend", "gloo",
    ] + list(extra))
    return len(run_training(args, mode))


def test_config1_resnet18_lp_224():
    # the exact BASELINE config-1 shape on gloo
    got = run_distributed(
        _run_mode, 2,
        ("lp",
         ("--model", "resnet18", "--image-size", "224", "--batch-size", "2",
          "--parts", "1", "--split-size", "2", "--num-classes", "1000")),
        timeout=300,
    )
    assert got[0] == 2


def test_config3_amoebanet_gems_8stages():
    # GEMS bidirectional pipeline over 8 stages (tiny model/filters)
    got = run_distributed(
        _run_mode, 8,
        ("gems",
         ("--model", "amoebanet", "--image-size", "64", "--batch-size", "8",
          "--parts", "2", "--split-size", "8", "--num-layers", "12",
          "--num-filters", "8", "--num-classes", "10")),
        timeout=600,
    )
    assert got[0] == 2


def test_config5_amoebanet_sp_gems_8ranks():
    """SP+GEMS+PP at the config-5 topology: 4 spatial tiles + LP stages,
    mp = 5 + 4 - 1 = 8 >= 2*4 tiles (mirror disjointness), two GEMS
    engines per rank. 8192^2 sizing is the GPU-scale variant; the
    schedule/seam/mirror wiring is what this validates."""
    got = run_distributed(
        _run_mode, 8,
        ("gems_sp",
         ("--model", "amoebanet", "--image-size", "128", "--batch-size", "4",
          "--parts", "2", "--split-size", "5", "--num-spatial-parts", "4",
          "--spatial-size", "1", "--slice-method", "vertical",
          "--num-layers", "6", "--num-filters", "8", "--num-classes", "10")),
        timeout=600,
    )
    assert got[0] == 2
