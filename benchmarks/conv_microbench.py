#!/usr/bin/env python3
"""Per-shape A/B of the gemscore implicit-GEMM MFMA conv vs MIOpen
(torch F.conv2d) on the shapes the flagship model actually runs.

Usage (GPU): python benchmarks/conv_microbench.py [--iters 20]
Writes a table to stdout; used to set the NativeConv2d dispatch policy.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

# (C, K, H, W, kh, kw, s, ph, pw) — amoebanet-D F416 @2048 + resnet shapes
SHAPES = [
    (3, 104, 2048, 2048, 3, 3, 2, 1, 1),     # stem
    (104, 208, 1024, 1024, 1, 1, 1, 0, 0),   # reduce 1x1 (big HW)
    (208, 52, 512, 512, 1, 7, 1, 0, 3),      # 1x7
    (208, 52, 512, 512, 7, 1, 1, 3, 0),      # 7x1
    (416, 104, 256, 256, 1, 7, 1, 0, 3),
    (416, 104, 256, 256, 3, 3, 1, 1, 1),     # conv_3x3 mid
    (64, 64, 512, 512, 3, 3, 1, 1, 1),       # resnet body
    (256, 256, 128, 128, 3, 3, 1, 1, 1),     # deeper
]


def timeit(fn, iters, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def winograd_ab(args):
    """F(2x2,3x3) batched-GEMM staging vs MIOpen at the bench 3x3 shapes."""
    from mpi4dl_amd.ops.winograd_ref import filter_transform, winograd_bmm_conv2d

    print(f"{'shape':<42} {'winograd':>9} {'miopen':>9} {'ratio':>6}")
    for C, K, H, W, kh, kw, s, ph, pw in SHAPES:
        if (kh, kw) != (3, 3) or s != 1:
            continue
        x = torch.randn(args.batch, C, H, W, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(K, C, 3, 3, device="cuda") * 0.05
        U = filter_transform(w)
        tw = timeit(lambda: winograd_bmm_conv2d(x, U, None, padding=ph),
                    args.iters)
        tm = timeit(
            lambda: torch.nn.functional.conv2d(
                x, w.to(torch.bfloat16), stride=1, padding=(ph, pw)
            ),
            args.iters,
        )
        y = winograd_bmm_conv2d(x, U, None, padding=ph)
        ref = torch.nn.functional.conv2d(
            x.float(), w, stride=1, padding=(ph, pw)
        )
        rel = (y - ref).abs().max().item() / max(ref.abs().max().item(), 1e-3)
        tag = f"C{C}->K{K} {H}x{W} k3x3"
        print(f"{tag:<42} {tw:8.3f}ms {tm:8.3f}ms {tm/tw:5.2f}x  relerr {rel:.3g}")


def model_pw_shapes(num_layers=18, num_filters=416, image=2048, mb=2):
    """Every distinct 1x1 conv shape the flagship model runs (meta pass)."""
    from mpi4dl_amd.models.amoebanet import amoebanetd
    from mpi4dl_amd.ops.conv_native import NativeConv2d

    shapes = {}
    orig = NativeConv2d.forward

    def probe(self, x):
        kh, kw = (
            (self.kernel_size, self.kernel_size)
            if isinstance(self.kernel_size, int)
            else self.kernel_size
        )
        if (kh, kw) == (1, 1):
            s = self.stride[0] if isinstance(self.stride, tuple) else self.stride
            key = (x.shape[1], self.out_channels, x.shape[2], x.shape[3], s)
            shapes[key] = shapes.get(key, 0) + 1
        return orig(self, x)

    NativeConv2d.forward = probe
    try:
        with torch.device("meta"):
            m = amoebanetd(1000, num_layers, num_filters)
            m(torch.zeros(mb, 3, image, image))
    finally:
        NativeConv2d.forward = orig
    return shapes


def pw_ab(args):
    """1x1 conv A/B: conv_pw streaming GEMM vs MIOpen, fwd + autograd,
    on the exact shapes of the flagship model."""
    from mpi4dl_amd.ops import backend
    from mpi4dl_amd.ops.conv_native import native_conv2d

    ge = backend.ext()
    shapes = model_pw_shapes(mb=args.batch)
    print(f"{'shape (xN uses)':<40} {'pw':>9} {'miopen':>9} {'ratio':>6}")
    tot_n = tot_m = 0.0
    for (C, K, H, W, s), uses in sorted(shapes.items()):
        x = torch.randn(args.batch, C, H, W, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(K, C, 1, 1, device="cuda", dtype=torch.bfloat16) * 0.05
        tn = timeit(lambda: ge.pw_fwd(x, w, None, s, s), args.iters)
        tm = timeit(
            lambda: torch.nn.functional.conv2d(x, w, stride=s), args.iters
        )
        y = ge.pw_fwd(x, w, None, s, s).float()
        ref = torch.nn.functional.conv2d(x.float(), w.float(), stride=s)
        rel = (y - ref).abs().max().item() / max(ref.abs().max().item(), 1e-3)
        tag = f"C{C}->K{K} {H}x{W} s{s} (x{uses})"
        print(f"{tag:<40} {tn:8.3f}ms {tm:8.3f}ms {tm/tn:5.2f}x  relerr {rel:.3g}",
              flush=True)
        tot_n += tn * uses
        tot_m += tm * uses
        # autograd legs
        xg = x.clone().requires_grad_(True)
        wf = w.float().requires_grad_(True)
        g = torch.randn_like(ref).to(torch.bfloat16)

        def nat_step():
            yy = native_conv2d(xg, wf, None, (s, s), (0, 0))
            yy.backward(g)
            xg.grad = None
            wf.grad = None

        x2 = x.clone().requires_grad_(True)
        w2 = w.float().requires_grad_(True)

        def ref_step():
            yy = torch.nn.functional.conv2d(
                x2, w2.to(torch.bfloat16), stride=s
            )
            yy.backward(g)
            x2.grad = None
            w2.grad = None

        # grad correctness
        yy = native_conv2d(xg, wf, None, (s, s), (0, 0))
        yy.backward(g)
        ry = torch.nn.functional.conv2d(x2.float(), w2, stride=s)
        ry.backward(g.float())
        relx = (xg.grad.float() - x2.grad).abs().max().item() / max(
            x2.grad.abs().max().item(), 1e-3
        )
        relw = (wf.grad - w2.grad).abs().max().item() / max(
            w2.grad.abs().max().item(), 1e-3
        )
        xg.grad = wf.grad = x2.grad = w2.grad = None
        tn2 = timeit(nat_step, max(args.iters // 2, 3))
        tm2 = timeit(ref_step, max(args.iters // 2, 3))
        print(f"{'':<40} f+b {tn2:8.3f}ms {tm2:8.3f}ms {tm2/tn2:5.2f}x  "
              f"relgx {relx:.3g} relgw {relw:.3g}", flush=True)
    print(f"\nuse-weighted fwd total: pw {tot_n:.2f}ms vs miopen {tot_m:.2f}ms "
          f"({tot_m/max(tot_n,1e-9):.2f}x)")


def bwdw_ab(args):
    """pw_bwdw vs torch conv2d_weight on the flagship 1x1 shapes."""
    from mpi4dl_amd.ops import backend

    ge = backend.ext()
    shapes = model_pw_shapes(mb=args.batch)
    print(f"{'shape (xN uses)':<40} {'pw_bwdw':>9} {'torch':>9} {'ratio':>6}")
    tot_n = tot_m = 0.0
    for (C, K, H, W, s), uses in sorted(shapes.items()):
        if s != 1:
            continue
        x = torch.randn(args.batch, C, H, W, device="cuda", dtype=torch.bfloat16)
        go = torch.randn(args.batch, K, H, W, device="cuda", dtype=torch.bfloat16)
        tn = timeit(lambda: ge.pw_bwdw(go, x), args.iters)
        tm = timeit(
            lambda: torch.nn.grad.conv2d_weight(
                x, (K, C, 1, 1), go, stride=1, padding=0
            ),
            args.iters,
        )
        gw = ge.pw_bwdw(go, x)
        ref = torch.nn.grad.conv2d_weight(
            x.float(), (K, C, 1, 1), go.float(), stride=1, padding=0
        ).view(K, C)
        rel = (gw - ref).abs().max().item() / max(ref.abs().max().item(), 1e-3)
        tag = f"C{C}->K{K} {H}x{W} (x{uses})"
        print(f"{tag:<40} {tn:8.3f}ms {tm:8.3f}ms {tm/tn:5.2f}x  relerr {rel:.3g}",
              flush=True)
        tot_n += tn * uses
        tot_m += tm * uses
    print(f"\nuse-weighted bwdw total: pw {tot_n:.2f}ms vs torch {tot_m:.2f}ms "
          f"({tot_m/max(tot_n,1e-9):.2f}x)")


def matmul_ab(args):
    """Fat 1x1 shapes: pw kernels vs a plain hipBLASLt matmul view
    (w[K,C] @ x[N,C,HW]) — the 'plain library GEMM' alternative."""
    from mpi4dl_amd.ops import backend

    ge = backend.ext()
    shapes = model_pw_shapes(mb=args.batch)
    print(f"{'shape (xN uses)':<40} {'pw':>9} {'matmul':>9} {'ratio':>6}"
          f" | {'bwdw':>9} {'einsum':>9} {'ratio':>6}")
    for (C, K, H, W, s), uses in sorted(shapes.items()):
        if s != 1:
            continue
        x = torch.randn(args.batch, C, H, W, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(K, C, device="cuda", dtype=torch.bfloat16) * 0.05
        w4 = w.view(K, C, 1, 1)
        xv = x.view(args.batch, C, H * W)
        tn = timeit(lambda: ge.pw_fwd(x, w4, None, 1, 1), args.iters)
        tm = timeit(lambda: torch.matmul(w, xv), args.iters)
        y = ge.pw_fwd(x, w4, None, 1, 1).float()
        ref = torch.matmul(w.float(), xv.float()).view(args.batch, K, H, W)
        rel = (y - ref).abs().max().item() / max(ref.abs().max().item(), 1e-3)
        # bwd-weight: native NT GEMM vs bmm+sum through hipBLASLt
        go = torch.randn(args.batch, K, H, W, device="cuda", dtype=torch.bfloat16)
        gv = go.view(args.batch, K, H * W)
        tb = timeit(lambda: ge.pw_bwdw(go, x), args.iters)
        te = timeit(
            lambda: torch.bmm(gv, xv.transpose(1, 2)).sum(dim=0), args.iters
        )
        tag = f"C{C}->K{K} {H}x{W} (x{uses})"
        print(f"{tag:<40} {tn:8.3f}ms {tm:8.3f}ms {tm/tn:5.2f}x"
              f" | {tb:8.3f}ms {te:8.3f}ms {te/tb:5.2f}x  relerr {rel:.3g}",
              flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--winograd", action="store_true",
                    help="A/B winograd_bmm_conv2d vs MIOpen on the 3x3 "
                         "stride-1 shapes (round-2 staging)")
    ap.add_argument("--pw", action="store_true",
                    help="A/B the conv_pw 1x1 kernel vs MIOpen on the "
                         "flagship model's 1x1 shapes")
    ap.add_argument("--bwdw", action="store_true",
                    help="A/B pw_bwdw vs torch conv2d_weight")
    ap.add_argument("--matmul", action="store_true",
                    help="A/B pw fat kernels vs a plain hipBLASLt matmul")
    args = ap.parse_args()
    if args.matmul:
        return matmul_ab(args)
    if args.winograd:
        return winograd_ab(args)
    if args.pw:
        return pw_ab(args)
    if args.bwdw:
        return bwdw_ab(args)
    from mpi4dl_amd.ops import backend

    ge = backend.ext()
    print(f"{'shape':<42} {'native':>9} {'miopen':>9} {'ratio':>6}")
    for C, K, H, W, kh, kw, s, ph, pw in SHAPES:
        x = torch.randn(args.batch, C, H, W, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(K, C, kh, kw, device="cuda", dtype=torch.bfloat16) * 0.05
        tn = timeit(lambda: ge.conv_fwd(x, w, None, s, s, ph, pw), args.iters)
        tm = timeit(
            lambda: torch.nn.functional.conv2d(x, w, stride=s, padding=(ph, pw)),
            args.iters,
        )
        # correctness spot-check
        y = ge.conv_fwd(x, w, None, s, s, ph, pw).float()
        ref = torch.nn.functional.conv2d(x, w, stride=s, padding=(ph, pw)).float()
        rel = (y - ref).abs().max().item() / max(ref.abs().max().item(), 1e-3)
        tag = f"C{C}->K{K} {H}x{W} k{kh}x{kw} s{s}"
        print(f"{tag:<42} {tn:8.3f}ms {tm:8.3f}ms {tm/tn:5.2f}x  relerr {rel:.3g}")
        # full autograd step (fwd+bwd) A/B
        if True:
            from mpi4dl_amd.ops.conv_native import native_conv2d

            wf = w.float().requires_grad_(True)
            xg = x.clone().requires_grad_(True)
            g = torch.randn_like(ref).to(torch.bfloat16)

            def nat_step():
                y = native_conv2d(xg, wf, None, (s, s), (ph, pw))
                y.backward(g)
                xg.grad = None
                wf.grad = None

            x2 = x.clone().requires_grad_(True)
            w2 = w.float().requires_grad_(True)

            def ref_step():
                y = torch.nn.functional.conv2d(x2, w2.to(torch.bfloat16),
                                               stride=s, padding=(ph, pw))
                y.backward(g)
                x2.grad = None
                w2.grad = None

            tn2 = timeit(nat_step, max(args.iters // 2, 3))
            tm2 = timeit(ref_step, max(args.iters // 2, 3))
            print(f"{'':<42} fwd+bwd {tn2:8.3f}ms {tm2:8.3f}ms {tm2/tn2:5.2f}x")


if __name__ == "__main__":
    main()
