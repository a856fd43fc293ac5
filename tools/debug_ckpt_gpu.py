"""GPU bisect for the act-ckpt gradient-parity failure (VERDICT weak #1).

Runs per-op micro checkpoint-parity tests for every native autograd
Function, a kernel-determinism probe, then the full failing-model repro
under native-op toggles (each toggle in a subprocess since the env
gates are read at dispatch time but MIOpen/extension state is global).

Usage (on a GPU box):  python tools/debug_ckpt_gpu.py [micro|model]
"""
from __future__ import annotations

import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.utils.checkpoint import checkpoint


def cmp_grads(tag, ga, gb, rtol=1e-2, atol=1e-3):
    ok = True
    for i, (x, y) in enumerate(zip(ga, gb)):
        if x is None and y is None:
            continue
        if not torch.allclose(x, y, rtol=rtol, atol=atol):
            d = (x - y).abs().max().item()
            m = x.abs().max().item()
            print(f"  [{tag}] grad {i} MISMATCH max|d|={d:.4f} max|x|={m:.4f}")
            ok = False
    print(f"[{tag}] {'OK' if ok else 'FAIL'}")
    return ok


def micro_op(tag, make_mod, xshape, autocast=True, dtype=torch.float32):
    """Grad parity: module(x) vs checkpoint(module, x)."""
    dev = torch.device("cuda", 0)

    def run(ck):
        torch.manual_seed(0)
        m = make_mod().to(dev)
        m.train()
        torch.manual_seed(1)
        x = torch.randn(*xshape, device=dev, dtype=dtype, requires_grad=True)
        ctxm = (
            torch.autocast("cuda", dtype=torch.bfloat16)
            if autocast
            else torch.autocast("cuda", enabled=False)
        )
        with ctxm:
            y = checkpoint(m, x, use_reentrant=False) if ck else m(x)
        loss = y.float().square().mean()
        loss.backward()
        gs = [x.grad.detach().float().clone()]
        gs += [
            p.grad.detach().float().clone() if p.grad is not None else None
            for p in m.parameters()
        ]
        return float(loss.detach()), gs

    la, ga = run(False)
    lb, gb = run(True)
    if abs(la - lb) > 1e-4 * max(abs(la), 1):
        print(f"  [{tag}] LOSS differs: {la} vs {lb}")
    return cmp_grads(tag, ga, gb)


def micros():
    from mpi4dl_amd.ops.conv_native import NativeConv2d
    from mpi4dl_amd.ops.norm import TileBatchNorm2d
    from mpi4dl_amd.ops.spatial_conv import HaloPool2d

    torch.cuda.init()
    results = {}
    # stem-shaped native conv (C=3 -> dispatches ConvFn)
    results["conv_stem_ac"] = micro_op(
        "conv_stem_ac",
        lambda: NativeConv2d(3, 16, 3, stride=2, padding=1, bias=False),
        (2, 3, 128, 128),
    )
    # 1x7 native conv path
    results["conv_1x7_ac"] = micro_op(
        "conv_1x7_ac",
        lambda: NativeConv2d(8, 8, (1, 7), padding=(0, 3), bias=False),
        (2, 8, 32, 512),
    )
    # fused BN (+relu) native
    results["bn_ac"] = micro_op(
        "bn_ac", lambda: TileBatchNorm2d(16), (2, 16, 64, 64)
    )

    def mk_bnrelu():
        bn = TileBatchNorm2d(16)
        bn.relu = True
        return bn

    results["bnrelu_ac"] = micro_op("bnrelu_ac", mk_bnrelu, (2, 16, 64, 64))
    results["maxpool_ac"] = micro_op(
        "maxpool_ac", lambda: HaloPool2d("max", 3, stride=1, padding=1), (2, 16, 64, 64)
    )
    results["avgpool_ac"] = micro_op(
        "avgpool_ac",
        lambda: HaloPool2d(
            "avg", 3, stride=1, padding=1, count_include_pad=False
        ),
        (2, 16, 64, 64),
    )
    # same without autocast (fp32 in, native BN/pool still dispatch)
    results["bn_fp32"] = micro_op(
        "bn_fp32", lambda: TileBatchNorm2d(16), (2, 16, 64, 64), autocast=False
    )
    results["maxpool_fp32"] = micro_op(
        "maxpool_fp32",
        lambda: HaloPool2d("max", 3, stride=1, padding=1),
        (2, 16, 64, 64),
        autocast=False,
    )

    # determinism probe: run each native op's fwd twice, compare bitwise
    from mpi4dl_amd.ops import backend

    ge = backend.ext()
    torch.manual_seed(0)
    x = torch.randn(2, 3, 128, 128, device="cuda").to(torch.bfloat16)
    w = torch.randn(16, 3, 3, 3, device="cuda").to(torch.bfloat16)
    y1 = ge.conv_fwd(x, w, None, 2, 2, 1, 1)
    y2 = ge.conv_fwd(x, w, None, 2, 2, 1, 1)
    print("conv_fwd deterministic:", bool((y1 == y2).all()))
    xb = torch.randn(2, 16, 64, 64, device="cuda").to(torch.bfloat16)
    s1 = ge.bn_stats64(xb)
    s2 = ge.bn_stats64(xb)
    print("bn_stats64 deterministic:", bool((s1 == s2).all()))
    print("RESULTS:", results)


def model_repro():
    """Full failing-test body; native toggles from env."""
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.amoebanet import amoebanetd
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=1)
    dev = torch.device("cuda", 0)

    def run(ckpt):
        torch.manual_seed(0)
        model = amoebanetd(100, 6, 64)
        gen = model_generator(model, 1, input_size=(2, 3, 256, 256))
        gen.get_output_shapes()
        gen.ready_model(0, device=dev)
        opt = torch.optim.SGD(gen.models.parameters(), lr=0.01, momentum=0.9)
        eng = train_model(
            gen, 0, 4, 2, comm, optimizer=opt, device=dev,
            autocast_dtype=torch.bfloat16, act_dtype=torch.bfloat16,
            act_ckpt=ckpt,
        )
        torch.manual_seed(42)
        x = torch.randn(4, 3, 256, 256, device=dev)
        y = torch.randint(0, 100, (4,), device=dev)
        torch.cuda.synchronize()
        loss, _, _ = eng.run_step(x, y)
        torch.cuda.synchronize()
        g = [p.grad.detach().float().clone() for p in gen.models.parameters()]
        return loss, g

    la, ga = run(False)
    lb, gb = run(True)
    print(f"loss {la:.6f} vs {lb:.6f}")
    bad = [
        i
        for i, (x, y) in enumerate(zip(ga, gb))
        if not torch.allclose(x, y, rtol=1e-2, atol=1e-3)
    ]
    env = {
        k: os.environ.get(k, "<unset>")
        for k in ("MPI4DL_NATIVE_CONV", "MPI4DL_NATIVE_BN", "MPI4DL_NATIVE_POOL")
    }
    print(f"MODEL {env}: bad_params={len(bad)}/{len(ga)} first_bad={bad[:6]}")


def main():
    mode = sys.argv[1] if len(sys.argv) > 1 else "all"
    if mode in ("micro", "all"):
        micros()
    if mode == "model":
        model_repro()
        return
    if mode == "all":
        base = dict(os.environ)
        for i, (conv, bn, pool) in enumerate([
            ("auto", "1", "1"),
            ("0", "1", "1"),
            ("auto", "0", "1"),
            ("auto", "1", "0"),
            ("0", "0", "0"),
        ]):
            env = dict(base)
            env.update(
                MPI4DL_NATIVE_CONV=conv, MPI4DL_NATIVE_BN=bn, MPI4DL_NATIVE_POOL=pool,
                MASTER_PORT=str(29640 + i),
            )
            print(f"--- model repro conv={conv} bn={bn} pool={pool} ---", flush=True)
            subprocess.run(
                [sys.executable, __file__, "model"], env=env, timeout=600
            )


if __name__ == "__main__":
    main()
