"""Shared benchmark runner for all entry points.

Reference parity: the per-benchmark main() bodies under
/root/reference/benchmarks/* (e.g. benchmark_resnet_sp.py:90-370) —
parse flags, build communicator + model + engine, loop epochs with
CUDA-event timing, print img/s on rank 0, loss/acc on the last rank.

Launch (one process per GPU over RCCL):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node <world> \
      --master-addr 127.0.0.1 benchmarks/<dir>/<script>.py [flags]
"""

from __future__ import annotations

import logging
import os
import statistics
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mpi4dl_amd import checkpoint as ckpt  # noqa: E402
from mpi4dl_amd.comm import Communicator, GradReducer  # noqa: E402
from mpi4dl_amd.ops.plan import SpatialPlan  # noqa: E402
from mpi4dl_amd.parallel.partition import model_generator  # noqa: E402
from mpi4dl_amd.parallel.pipeline import train_model  # noqa: E402
from mpi4dl_amd.parallel.spatial import (  # noqa: E402
    train_model_spatial,
    verify_spatial_config,
)
from mpi4dl_amd.parser import get_parser, parse_csv_ints  # noqa: E402
from mpi4dl_amd.utils import GLOBAL_TIMER  # noqa: E402

log = logging.getLogger("mpi4dl_amd.benchmark")


def build_model(args, plan, mb):
    shape = (mb, 3, args.image_size, args.image_size)
    torch.manual_seed(0)
    if args.model == "resnet18":
        # BASELINE config 1: ResNet-18 layer parallelism (no spatial variant
        # — LP/PP/GEMS/DP only)
        assert plan is None, "resnet18 is LP-only; use resnet/resnet101 for SP"
        from mpi4dl_amd.models.resnet import get_resnet18_cells

        return get_resnet18_cells(shape, args.num_classes)
    if args.model == "resnet101":
        # BASELINE config 4: ResNet-101 SP+PP at high resolution
        from mpi4dl_amd.models import resnet_spatial as M

        return M.get_resnet101_cells(shape, args.num_classes, plan=plan)
    if args.model == "resnet":
        if args.halo_d2:
            from mpi4dl_amd.models import resnet_spatial_d2 as M

            return M.get_resnet_v2(
                shape, args.num_classes, n=max(args.num_layers // 9, 1),
                num_filters=args.num_filters if args.num_filters <= 64 else 16,
                plan=plan, fused_layers=args.fused_layers,
                ref_stem=args.ref_stem,
            )
        from mpi4dl_amd.models import resnet_spatial as M

        return M.get_resnet_v2(
            shape, args.num_classes, n=max(args.num_layers // 9, 1),
            num_filters=args.num_filters if args.num_filters <= 64 else 16,
            plan=plan, ref_stem=args.ref_stem,
        )
    from mpi4dl_amd.models.amoebanet import amoebanetd

    layers = args.num_layers - (args.num_layers % 3) or 3
    return amoebanetd(args.num_classes, layers, args.num_filters, plan=plan,
                      ref_quirks=getattr(args, "ref_quirks", False))


def make_engines(args, mode):
    """mode: 'lp' | 'sp' | 'gems' | 'gems_sp'. Returns (step_fn, comm, extras)."""
    logging.basicConfig(level=logging.DEBUG if args.verbose else logging.INFO)
    spatial = mode in ("sp", "gems_sp")
    gems = mode in ("gems", "gems_sp")
    nsp = parse_csv_ints(args.num_spatial_parts)
    if spatial:
        verify_spatial_config(args.slice_method, args.image_size, nsp)
    comm = Communicator(
        split_size=args.split_size,
        ENABLE_SPATIAL=spatial,
        num_spatial_parts=nsp if spatial else 1,
        spatial_size=args.spatial_size if spatial else 0,
        LOCAL_DP_LP=args.local_DP,
        ENABLE_GEMS=gems,
        backend=args.backend,
    )
    on_gpu = torch.cuda.is_available()
    device = (
        torch.device("cuda", torch.cuda.current_device())
        if on_gpu
        else torch.device("cpu")
    )
    mb = args.batch_size // args.parts
    balance = parse_csv_ints(args.balance)

    def build_gen(plan, pos):
        model = build_model(args, plan, mb)
        gen = model_generator(
            model,
            args.split_size,
            (mb, 3, args.image_size, args.image_size),
            balance=balance,
        )
        gen.get_output_shapes()
        gen.ready_model(comm.get_split_rank(pos), device=device)
        return gen

    def mkopt(module):
        name = getattr(args, "optimizer", "sgd").lower()
        if name == "adam":
            return torch.optim.Adam(
                module.parameters(), lr=args.learning_rate,
                weight_decay=args.weight_decay,
            )
        if name == "adamw":
            return torch.optim.AdamW(
                module.parameters(), lr=args.learning_rate,
                weight_decay=args.weight_decay,
            )
        if name != "sgd":
            raise ValueError(f"--optimizer {name}: use sgd / adam / adamw")
        if on_gpu:
            from mpi4dl_amd.optim import FusedSGD

            return FusedSGD(module, lr=args.learning_rate,
                            momentum=args.momentum,
                            weight_decay=args.weight_decay)
        return torch.optim.SGD(
            module.parameters(), lr=args.learning_rate,
            momentum=args.momentum, weight_decay=args.weight_decay,
        )

    eng_kw = dict(
        device=device,
        autocast_dtype=torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else None,
        act_dtype=torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else torch.float32,
        schedule=getattr(args, "schedule", "gpipe"),
        act_ckpt=getattr(args, "act_ckpt", False),
    )
    reducer = GradReducer(comm, fp16_allreduce=args.fp16_allreduce)

    if not gems:
        plan = (
            SpatialPlan(comm, _resolve_balance(args, comm, mb), args.slice_method,
                        grad_mode=args.grad_mode)
            if spatial
            else None
        )
        gen = build_gen(plan, comm.local_rank)
        opt = mkopt(gen.models)
        if spatial:
            eng = train_model_spatial(
                gen, comm.local_rank, args.batch_size, args.parts, comm,
                slice_method=args.slice_method, optimizer=opt,
                grad_mode=args.grad_mode, **eng_kw,
            )
        else:
            eng = train_model(
                gen, comm.local_rank, args.batch_size, args.parts, comm,
                optimizer=opt, **eng_kw,
            )

        # bucketed allreduce-during-backward on the primary gradient group
        # (default on; reference equivalent: DDP wrap mp_pipeline.py:92-124)
        overlap = (
            reducer.setup_overlap(eng.models)
            if not getattr(args, "no_overlap_grads", False)
            else None
        )

        def step(x, y):
            loss, corr, seen = eng.run_step(x, y)
            if overlap is not None:
                reducer.finish_overlap(eng.models)
                reducer.apply_allreduce(eng.models, skip_group=overlap["group"])
            else:
                reducer.apply_allreduce(eng.models)
            eng.update()
            return loss, corr, seen

        return step, comm, {"engine": eng, "gen": gen, "optimizer": opt,
                            "batch_per_step": args.batch_size}

    # GEMS modes
    from mpi4dl_amd.parallel.gems import (
        train_model_master,
        train_spatial_model_master,
    )

    r = comm.rank % comm.mp_size
    if mode == "gems":
        gen1 = build_gen(None, r)
        gen2 = build_gen(None, comm.mp_size - 1 - r)
        eng = train_model_master(
            gen1, gen2, args.batch_size, args.parts, comm,
            replications=args.times,
            enable_comm_opt=args.enable_master_comm_opt,
            fp16_allreduce=args.fp16_allreduce,
            optimizer=mkopt(gen1.models), **eng_kw,
        )
        eng.train_model2.optimizer = mkopt(gen2.models)
    else:
        bal = _resolve_balance(args, comm, mb)
        plan1 = SpatialPlan(comm, bal, args.slice_method, grad_mode=args.grad_mode)
        plan2 = SpatialPlan(comm, bal, args.slice_method, grad_mode=args.grad_mode,
                            gems_inverse=True)
        gen1 = build_gen(plan1, r)
        gen2 = build_gen(plan2, comm.mp_size - 1 - r)
        eng = train_spatial_model_master(
            gen1, gen2, args.batch_size, args.parts, comm,
            slice_method=args.slice_method, replications=args.times,
            enable_comm_opt=args.enable_master_comm_opt,
            fp16_allreduce=args.fp16_allreduce,
            optimizer=mkopt(gen1.models), **eng_kw,
        )
        eng.train_model2.optimizer = mkopt(gen2.models)

    def step(x, y):
        loss, corr, seen = eng.run_step(x, y)
        eng.allreduce_and_update()
        return loss, corr, seen

    return step, comm, {
        "engine": eng,
        "gen": gen1,
        "gen2": gen2,
        "optimizer": eng.train_model1.optimizer,
        "optimizer2": eng.train_model2.optimizer,
        "batch_per_step": 2 * args.times * args.batch_size,
    }


def _resolve_balance(args, comm, mb):
    bal = parse_csv_ints(args.balance)
    if bal is not None:
        return bal
    # probe cell count and split evenly (meta, free)
    with torch.device("meta"):
        ncells = len(build_model(args, None, mb))
    base, rem = divmod(ncells, args.split_size)
    return [base + (1 if i < rem else 0) for i in range(args.split_size)]


def run_training(args, mode):
    step, comm, extras = make_engines(args, mode)
    on_gpu = torch.cuda.is_available()
    B = extras["batch_per_step"]
    torch.manual_seed(1405 + comm.rank)  # reference seeds a fixed value too
    S = args.image_size

    loader = None
    if args.app in (1, 2):
        # real data: ImageFolder / CIFAR-10 (reference APP wiring,
        # benchmark_resnet_lp.py:183-208); every rank iterates the same
        # deterministic batch sequence
        from mpi4dl_amd.data import make_dataloader

        loader, n = make_dataloader(
            args.app, args.datapath, B, S, args.num_classes,
            num_workers=args.num_workers,
        )
        log.info("dataset: app=%d size=%d (%d batches/epoch)", args.app, n,
                 len(loader))
        batches = iter(loader)
    else:
        # synthetic: one fixed random batch (perf benchmarking path)
        x = torch.randn(B, 3, S, S)
        y = torch.randint(0, args.num_classes, (B,))
        if on_gpu:
            x, y = x.cuda(), y.cuda()

    if args.resume and args.checkpoint_dir:
        gen2 = extras.get("gen2")
        ckpt.load_checkpoint(
            args.checkpoint_dir, extras["gen"].models, extras.get("optimizer"),
            comm, module2=gen2.models if gen2 is not None else None,
            optimizer2=extras.get("optimizer2"),
        )
        eng = extras["engine"]
        if gen2 is None and hasattr(eng, "sync_models"):
            # old single-replica checkpoint: mirror replica 1 onto 2
            eng.sync_models()

    times = []
    for epoch in range(args.num_epochs):
        for it in range(args.num_steps):
            if loader is not None:
                try:
                    x, y = next(batches)
                except StopIteration:
                    batches = iter(loader)
                    x, y = next(batches)
                if on_gpu:
                    x, y = x.cuda(non_blocking=True), y.cuda(non_blocking=True)
            t0 = time.perf_counter()
            loss, corr, seen = step(x, y)
            if on_gpu:
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            times.append(dt)
            if comm.rank == comm.world_size - 1 and seen:
                log.info(
                    "epoch %d step %d loss %.4f acc %.3f %.1f ms",
                    epoch, it, loss, corr / max(seen, 1), dt * 1e3,
                )
    if args.checkpoint_dir and not args.resume:
        gen2 = extras.get("gen2")
        ckpt.save_checkpoint(
            args.checkpoint_dir, extras["gen"].models, extras.get("optimizer"),
            comm, module2=gen2.models if gen2 is not None else None,
            optimizer2=extras.get("optimizer2"),
        )

    if args.enable_evaluation:
        # forward-only pass over held-out batches (CIFAR test split when
        # --app 2, else fresh synthetic data); top-1 acc on the last rank
        eng = extras["engine"]
        Be = args.batch_size  # GEMS evals one replica: plain batch, not 2x
        if args.app == 2:
            from mpi4dl_amd.data import make_dataloader

            try:
                ev_loader, _ = make_dataloader(
                    2, args.datapath, Be, S, args.num_classes,
                    num_workers=args.num_workers, train=False,
                )
            except FileNotFoundError:
                log.warning("no CIFAR test_batch on disk; evaluating on "
                            "the train split")
                ev_loader, _ = make_dataloader(
                    2, args.datapath, Be, S, args.num_classes,
                    num_workers=args.num_workers, train=True,
                )
            ev_batches = list(ev_loader)[: max(args.num_steps, 1)]
        else:
            g = torch.Generator().manual_seed(7)
            ev_batches = [
                (torch.randn(Be, 3, S, S, generator=g),
                 torch.randint(0, args.num_classes, (Be,), generator=g))
                for _ in range(max(args.num_steps, 1))
            ]
        tot_loss, tot_corr, tot_seen = 0.0, 0, 0
        for x, y in ev_batches:
            if on_gpu:
                x, y = x.cuda(), y.cuda()
            loss, corr, seen = eng.run_eval(x, y)
            tot_loss += loss
            tot_corr += corr
            tot_seen += seen
        if comm.rank == comm.world_size - 1 and tot_seen:
            print(
                f"Eval loss {tot_loss / len(ev_batches):.4f} "
                f"acc {tot_corr / tot_seen:.3f} over {tot_seen} samples"
            )
    if comm.rank == 0:
        steady = times[1:] or times
        img_s = B / statistics.median(steady)
        print(
            f"Mean {B / (sum(steady) / len(steady)):.3f} img/s "
            f"Median {img_s:.3f} img/s over {len(times)} steps "
            f"({statistics.median(steady) * 1e3:.1f} ms/step)"
        )
        if GLOBAL_TIMER.totals:
            print(GLOBAL_TIMER.report())
    return times


def main(mode):
    args = get_parser().parse_args()
    run_training(args, mode)
