#!/usr/bin/env python3
"""amoebanet with mode 'gems_sp' (reference parity: benchmarks/gems_master_with_spatial_parallelism/benchmark_amoebanet_gems_master_with_sp.py).

Launch: python -m torch.distributed.run --nnodes=1 --nproc-per-node <N> \
    --master-addr 127.0.0.1 benchmarks/gems_master_with_spatial_parallelism/benchmark_amoebanet_gems_master_with_sp.py [flags]
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
sys.argv += ["--model", "amoebanet"] if "--model" not in " ".join(sys.argv) else []
from runner import main  # noqa: E402

if __name__ == "__main__":
    main("gems_sp")
