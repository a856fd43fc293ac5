#!/usr/bin/env python3
"""Thin entry (reference parity: benchmarks/communication/halo/benchmark_sp_halo_exchange_with_compute.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.argv += ["--mode", "compute"]
from halo_bench import main  # noqa: E402

if __name__ == "__main__":
    main()
