#!/usr/bin/env python3
"""Halo-exchange micro-benchmarks + exact validation.

Reference parity: benchmarks/communication/halo/benchmark_sp_halo_exchange*.py
(exchange-only :90-613, with-compute, and full conv_spatial validation
against the undistributed conv). Ground truth is the reference's
integer-arange oracle (create_input_* :417-566, test_output :568-578).

Launch: python -m torch.distributed.run --nnodes=1 --nproc-per-node <N> \
    --master-addr 127.0.0.1 benchmarks/communication/halo/halo_bench.py \
    --mode exchange|compute|conv --image-size 1024 --halo-len 3 \
    --slice-method vertical --iterations 100 --warmup 10
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(
    0,
    os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."),
)

from mpi4dl_amd.comm import init_distributed  # noqa: E402
from mpi4dl_amd.ops.halo import HaloExchanger, TileLayout, halo_pad  # noqa: E402
from mpi4dl_amd.ops.spatial_conv import HaloConv2d  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="exchange",
                    choices=["exchange", "compute", "conv"])
    ap.add_argument("--image-size", type=int, default=1024)
    ap.add_argument("--halo-len", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=1)
    ap.add_argument("--channels", type=int, default=3)
    ap.add_argument("--out-channels", type=int, default=256)
    ap.add_argument("--slice-method", default="vertical",
                    choices=["square", "vertical", "horizontal"])
    ap.add_argument("--iterations", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--validate", action="store_true", default=True)
    args = ap.parse_args()

    import torch.distributed as dist

    init_distributed()
    rank, world = dist.get_rank(), dist.get_world_size()
    dev = (
        torch.device("cuda", torch.cuda.current_device())
        if torch.cuda.is_available()
        else torch.device("cpu")
    )
    layout = TileLayout(world, args.slice_method)
    S, h = args.image_size, args.halo_len
    B, C = args.batch_size, args.channels

    # integer arange image -> exact-match oracle (reference :417)
    full = (
        torch.arange(B * C * S * S, dtype=torch.float32).reshape(B, C, S, S)
        % 997.0
    ).to(dev)
    tile = layout.slice_input(full, rank).contiguous()
    ex = HaloExchanger(layout, rank, lambda t: t)

    k = 2 * h + 1
    if args.mode == "conv":
        torch.manual_seed(0)
        conv = HaloConv2d(
            C, args.out_channels, k, num_spatial_parts=world,
            slice_method=args.slice_method, spatial_local_rank=rank,
        ).to(dev)
        ref = torch.nn.Conv2d(C, args.out_channels, k, padding=h).to(dev)
        with torch.no_grad():
            ref.weight.copy_(conv.conv.weight)
            ref.bias.copy_(conv.conv.bias)
        run = lambda: conv(tile)
    elif args.mode == "compute":
        torch.manual_seed(0)
        w = torch.randn(args.out_channels, C, k, k, device=dev)

        def run():
            xp = halo_pad(tile, h, ex)
            return F.conv2d(xp, w)
    else:
        run = lambda: halo_pad(tile, h, ex)

    # validation (exact: the halo'd tile must equal a slice of the padded
    # full image; conv mode validates against the undistributed conv)
    if args.validate:
        with torch.no_grad():
            out = run()
            if args.mode == "exchange":
                fullp = F.pad(full, (h, h, h, h))
                r, c = layout.pos(rank)
                th, tw = S // layout.rows, S // layout.cols
                expect = fullp[
                    :, :, r * th : r * th + th + 2 * h,
                    c * tw : c * tw + tw + 2 * h,
                ]
                ok = torch.equal(out, expect)
            elif args.mode == "conv":
                expect = layout.slice_input(ref(full), rank)
                ok = torch.allclose(out.float(), expect, atol=1e-3)
            else:
                ok = torch.isfinite(out).all()
        print(f"Validation {'passed' if ok else 'FAILED'} Rank:{rank}", flush=True)
        assert ok

    # timing (reference :598-613)
    with torch.no_grad():
        for _ in range(args.warmup):
            run()
        if dev.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.iterations):
            run()
        if dev.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        dt = (time.perf_counter() - t0) / args.iterations
    if rank == 0:
        print(
            f"mode={args.mode} image={S} halo={h} parts={world} "
            f"({args.slice_method}): {dt * 1e3:.3f} ms/iter",
            flush=True,
        )


if __name__ == "__main__":
    main()
