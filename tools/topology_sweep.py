#!/usr/bin/env python3
"""Randomized (seeded) SP/LP topology parity sweep on gloo.

Samples valid (slice_method, nsp, spatial_size, split, parts, schedule,
grad_mode, local_DP) combinations, trains each 2 steps distributed and
serially, and checks trajectory parity. A pre-release brute-force over
the seam space — run manually:

    python tools/topology_sweep.py [--combos N] [--seed S]

(Not part of the pytest suite: each combo spawns a process group; the
suite keeps the curated parity subset.)
"""
import argparse
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(
    0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests")
)

import torch  # noqa: E402

from dist_util import run_distributed  # noqa: E402

IMG = 32
NCLS = 10


def serial_losses(steps, batch, parts, lr):
    from mpi4dl_amd.models.resnet import get_resnet_v1

    torch.manual_seed(0)
    model = get_resnet_v1((batch, 3, IMG, IMG), num_classes=NCLS, n=1, num_filters=8)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    crit = torch.nn.CrossEntropyLoss()
    torch.manual_seed(42)
    out = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        tot = 0.0
        for px, py in zip(x.chunk(parts), y.chunk(parts)):
            loss = crit(model(px).float(), py)
            (loss / parts).backward()
            tot += float(loss.detach())
        opt.step()
        opt.zero_grad(set_to_none=False)
        out.append(tot / parts)
    return out


def body(rank, world, steps, batch, parts, lr, cfg):
    from mpi4dl_amd.comm import Communicator, GradReducer
    from mpi4dl_amd.models import resnet_spatial
    from mpi4dl_amd.ops.plan import SpatialPlan
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.spatial import train_model_spatial

    comm = Communicator(
        split_size=cfg["split"], ENABLE_SPATIAL=True,
        num_spatial_parts=cfg["nsp"], spatial_size=cfg["spatial_size"],
        LOCAL_DP_LP=cfg["ldp"], backend="gloo",
    )
    torch.manual_seed(0)
    probe = resnet_spatial.get_resnet_v1((1, 3, IMG, IMG), NCLS, n=1, num_filters=8)
    ncells = len(probe)
    base, rem = divmod(ncells, cfg["split"])
    balance = [base + (1 if i < rem else 0) for i in range(cfg["split"])]
    plan = SpatialPlan(comm, balance, cfg["slice"], grad_mode=cfg["grad_mode"])
    torch.manual_seed(0)
    model = resnet_spatial.get_resnet_v1(
        (batch // parts, 3, IMG, IMG), NCLS, n=1, num_filters=8, plan=plan
    )
    gen = model_generator(model, cfg["split"],
                          input_size=(batch // parts, 3, IMG, IMG), balance=balance)
    gen.get_output_shapes()
    gen.ready_model(comm.split_rank, device=torch.device("cpu"))
    opt = torch.optim.SGD(gen.models.parameters(), lr=lr, momentum=0.9)
    eng = train_model_spatial(
        gen, comm.local_rank, batch, parts, comm, slice_method=cfg["slice"],
        optimizer=opt, grad_mode=cfg["grad_mode"], schedule=cfg["schedule"],
        act_ckpt=cfg.get("act_ckpt", False),
        device=torch.device("cpu"),
    )
    red = GradReducer(comm)
    torch.manual_seed(42)
    losses = []
    for _ in range(steps):
        x = torch.randn(batch, 3, IMG, IMG)
        y = torch.randint(0, NCLS, (batch,))
        loss, _, _ = eng.run_step(x, y)
        red.apply_allreduce(eng.models)
        eng.update()
        losses.append(loss)
    return losses


def sample_cfg(rng):
    slice_ = rng.choice(["vertical", "horizontal", "square"])
    nsp_choices = [4] if slice_ == "square" else [2, [4, 2]]
    nsp = rng.choice(nsp_choices)
    spatial_size = 2 if isinstance(nsp, list) else 1
    ldp = rng.choice([1, 1, 2])
    split = spatial_size + rng.choice([1, 2])
    return dict(slice=slice_, nsp=nsp, spatial_size=spatial_size,
                split=split, ldp=ldp,
                schedule=rng.choice(["gpipe", "1f1b"]),
                act_ckpt=rng.random() < 0.3,
                grad_mode=rng.choice(["exact", "exact", "drop"]))


def world_of(cfg):
    from mpi4dl_amd.comm import compute_mp_size

    return compute_mp_size(cfg["split"], cfg["nsp"], cfg["spatial_size"],
                           cfg["ldp"])


def gems_sweep(args, rng):
    """GEMS LP sweep: split counts (odd/even mp), parts, MASTER-OPT."""
    sys.path.insert(0, os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
    import test_gems as TG

    steps, lr = 2, 0.01
    failures = 0
    for i in range(args.combos):
        split = rng.choice([2, 3, 4])
        parts = rng.choice([1, 2])
        comm_opt = rng.random() < 0.5
        B = 2 * parts
        tag = f"[gems {i}] split={split} parts={parts} B={B} comm_opt={comm_opt}"
        try:
            expected = TG._serial_losses(steps, B, parts, lr)
            got = run_distributed(TG._gems_body, split,
                                  (steps, B, parts, lr, comm_opt), timeout=300)
            combined = [sum(g[s] for g in got) for s in range(steps)]
            errs = [abs(e - g) for e, g in zip(expected, combined)]
            ok = max(errs) < 5e-4
            print(f"{tag}  ->  {'OK' if ok else 'FAIL'} maxerr={max(errs):.2e}")
            failures += 0 if ok else 1
        except Exception as e:  # noqa: BLE001
            print(f"{tag}  ->  ERROR {type(e).__name__}: {str(e)[:200]}")
            failures += 1
    print(f"\n{args.combos - failures}/{args.combos} gems passed")
    sys.exit(1 if failures else 0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--combos", type=int, default=8)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--gems", action="store_true",
                    help="sweep GEMS LP configs instead of SP/LP seams")
    args = ap.parse_args()
    if args.gems:
        return gems_sweep(args, random.Random(args.seed))
    rng = random.Random(args.seed)
    steps, lr = 2, 0.01
    failures = 0
    for i in range(args.combos):
        cfg = sample_cfg(rng)
        world = world_of(cfg)
        # batch: divisible by parts and by local-DP shards
        parts = rng.choice([1, 2])
        batch = parts * cfg["ldp"] * 2
        tag = (f"[{i}] {cfg['slice']} nsp={cfg['nsp']} ss={cfg['spatial_size']} "
               f"split={cfg['split']} ldp={cfg['ldp']} parts={parts} "
               f"{cfg['schedule']}/{cfg['grad_mode']}"
               f"{'/ckpt' if cfg.get('act_ckpt') else ''} world={world}")
        try:
            expected = serial_losses(steps, batch, parts, lr)
            got = run_distributed(body, world, (steps, batch, parts, lr, cfg),
                                  timeout=300)
            if cfg["ldp"] > 1:
                # local-DP last stages report shard losses; average them
                L = cfg["ldp"]
                last = got[-L:]
                final = [sum(r[s] for r in last) / L for s in range(steps)]
            else:
                final = got[-1]
            errs = [abs(e - g) for e, g in zip(expected, final)]
            ok = (max(errs) < 5e-4) if cfg["grad_mode"] == "exact" else all(
                abs(g) < 1e3 for g in final)
            print(f"{tag}  ->  {'OK' if ok else 'FAIL'} maxerr={max(errs):.2e}")
            if not ok:
                failures += 1
        except Exception as e:  # noqa: BLE001
            print(f"{tag}  ->  ERROR {type(e).__name__}: {str(e)[:200]}")
            failures += 1
    print(f"\n{args.combos - failures}/{args.combos} passed")
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
