"""Shared CLI flag system (reference parity: src/torchgems/parser.py:21-143).

All benchmark entry points share this parser; flag names match the
reference so existing launch scripts translate 1:1. Launch is
torchrun-style env rendezvous (one process per GPU over RCCL) instead
of mpirun_rsh.
"""

from __future__ import annotations

import argparse


def get_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        description="mpi4dl_amd benchmarks (MI355X-native MPI4DL)",
        formatter_class=argparse.ArgumentDefaultsHelpFormatter,
    )
    p.add_argument("--fp16-allreduce", action="store_true",
                   help="reduce gradients as bf16 (half the xGMI bytes, "
                        "fp32 exponent range kept); applies to all engines "
                        "including GEMS")
    p.add_argument("--no-overlap-grads", action="store_true",
                   help="disable bucketed allreduce-during-backward on the "
                        "primary gradient group (overlap is the default)")
    p.add_argument("--model", default="resnet",
                   choices=["resnet", "resnet18", "resnet101", "amoebanet"])
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--parts", type=int, default=4,
                   help="pipeline micro-batches per step")
    p.add_argument("--split-size", type=int, default=2,
                   help="number of LP/pipeline partitions")
    p.add_argument("--num-spatial-parts", type=str, default="4",
                   help="csv: tiles per spatial partition")
    p.add_argument("--spatial-size", type=int, default=1,
                   help="leading partitions that are spatial")
    p.add_argument("--times", type=int, default=1,
                   help="GEMS replication pairs per step")
    p.add_argument("--image-size", type=int, default=1024)
    p.add_argument("--num-epochs", type=int, default=1)
    p.add_argument("--num-steps", type=int, default=8,
                   help="steps per epoch on synthetic data")
    p.add_argument("--num-layers", type=int, default=18)
    p.add_argument("--num-filters", type=int, default=416)
    p.add_argument("--num-classes", type=int, default=10)
    p.add_argument("--balance", type=str, default=None,
                   help="csv: cells per partition")
    p.add_argument("--halo-d2", "--halo-D2", action="store_true",
                   dest="halo_d2",
                   help="use the D2 fused-halo model variant "
                        "(--halo-D2 = the reference's spelling)")
    p.add_argument("--fused-layers", type=int, default=4,
                   help="blocks per fused D2 halo exchange")
    p.add_argument("--local-DP", type=int, default=1,
                   help="LBANN-style local DP inside LP partitions")
    p.add_argument("--slice-method", default="square",
                   choices=["square", "vertical", "horizontal"])
    p.add_argument("--app", type=int, default=3,
                   help="1=medical,2=cifar,3=synthetic (only 3 is bundled)")
    p.add_argument("--datapath", default=None)
    p.add_argument("--enable-master-comm-opt", action="store_true",
                   help="GEMS MASTER-OPT overlapped grad swap")
    p.add_argument("--num-gpus-mp", type=int, default=1,
                   help="compat flag (topology comes from WORLD_SIZE)")
    p.add_argument("--num-workers", type=int, default=0)
    p.add_argument("--optimizer", default="sgd", choices=["sgd", "adam", "adamw"])
    p.add_argument("--learning-rate", type=float, default=0.001)
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--schedule", default="gpipe", choices=["gpipe", "1f1b"])
    p.add_argument("--grad-mode", default="exact", choices=["exact", "drop"],
                   help="halo backward: exact transposed exchange or the "
                        "reference's drop semantics")
    p.add_argument("--ref-quirks", action="store_true",
                   help="reproduce the reference's AmoebaNet bitwise (its "
                        "max_pool_3x3 builder creates an AvgPool)")
    p.add_argument("--ref-stem", action="store_true",
                   help="use the reference's stride-1 3x3 stem at any image "
                        "size (for apples-to-apples A/B with its published "
                        "ResNet numbers; default stem downsamples >=128^2)")
    p.add_argument("--act-ckpt", action="store_true",
                   help="activation checkpointing: recompute cell forwards "
                        "in backward (cuts GPipe peak activation memory)")
    p.add_argument("--checkpoint-dir", default=None)
    p.add_argument("--resume", action="store_true")
    p.add_argument("--verbose", action="store_true")
    p.add_argument("--enable-evaluation", action="store_true")
    p.add_argument("--backend", default=None, help="nccl/gloo override")
    return p


def parse_csv_ints(s):
    if s is None:
        return None
    return [int(v) for v in str(s).split(",") if v != ""]
