#!/usr/bin/env python3
"""Distributed training entry point.

Reference parity: `upstream:dist_trainer.py` with
1:1 flag names (--dnn --dataset --batch-size --nsteps-update --lr
--max-epochs --density --compression ...).  Launch model is
MI355X-native: one process per GPU via torchrun + RCCL instead of
mpirun + mpi4py, e.g.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --dnn resnet50 \
        --dataset imagenet --density 0.001

Single-process runs work without torchrun (world_size = 1).
"""
import argparse
import os
import time

import torch
import torch.distributed as dist

import gtopkssgd_amd as gt
from gtopkssgd_amd.settings import logger
from gtopkssgd_amd.trainer import DLTrainer
from gtopkssgd_amd.utils import JsonlLogger, gen_random_id


def parse_args():
    p = argparse.ArgumentParser(description="gTop-k S-SGD (MI355X-native)")
    p.add_argument("--dnn", default="resnet20")
    p.add_argument("--dataset", default="cifar10",
                   choices=["mnist", "cifar10", "imagenet", "ptb", "an4",
                            "an4wav"])
    p.add_argument("--data-dir", default="./data", help="unused (synthetic)")
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--nsteps-update", type=int, default=1)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight-decay", type=float, default=5e-4)
    p.add_argument("--max-epochs", type=int, default=1)
    p.add_argument("--iters-per-epoch", type=int, default=0,
                   help="0 = full pass over the (synthetic) dataset")
    p.add_argument("--density", type=float, default=0.001)
    p.add_argument("--compression", default="gtopk",
                   choices=["gtopk", "topk", "none"])
    p.add_argument("--nworkers", type=int, default=1,
                   help="informational; actual world size comes from env")
    p.add_argument("--nwpernode", type=int, default=1)
    p.add_argument("--num-workers", type=int, default=2)
    p.add_argument("--n-samples", type=int, default=None)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--device", default=None)
    p.add_argument("--no-amp", action="store_true")
    p.add_argument("--norm-clip", type=float, default=None)
    p.add_argument("--checkpoint-dir", default="./checkpoints")
    p.add_argument("--run-tag", default=None)
    p.add_argument("--resume", action="store_true",
                   help="resume from the latest checkpoint of this run-tag "
                        "(model + optimizer + EF residual state)")
    p.add_argument("--metrics", default=None, help="JSONL metrics path")
    p.add_argument("--stats", default=None,
                   help="JSONL per-phase optimizer stats path (rank 0)")
    p.add_argument("--profile", action="store_true",
                   help="enable HIP-event phase timers in the optimizer")
    p.add_argument("--torch-profile", default=None, metavar="DIR",
                   help="capture a torch.profiler chrome trace of a few "
                        "steps into DIR (rank 0)")
    return p.parse_args()


def init_dist(args):
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        use_gpu = torch.cuda.is_available() and args.device != "cpu"
        backend = "nccl" if use_gpu else "gloo"
        dist.init_process_group(backend=backend)
        rank = dist.get_rank()
        world = dist.get_world_size()
        if use_gpu:
            # torchrun exports LOCAL_RANK; fall back to packed layout
            local = int(os.environ.get(
                "LOCAL_RANK", rank % torch.cuda.device_count()))
            torch.cuda.set_device(local)
        return rank, world
    return 0, 1


def main():
    args = parse_args()
    rank, world = init_dist(args)
    run_tag = args.run_tag or (
        f"{args.dnn}-{args.dataset}-bs{args.batch_size}-"
        f"rho{args.density}-{args.compression}-n{world}")
    metrics = JsonlLogger(args.metrics) if args.metrics and rank == 0 else None

    trainer = DLTrainer(
        rank=rank, size=world, dnn=args.dnn, dataset=args.dataset,
        batch_size=args.batch_size, nsteps_update=args.nsteps_update,
        lr=args.lr, device=args.device, amp=not args.no_amp,
        num_workers=args.num_workers, n_samples=args.n_samples,
        seed=args.seed, norm_clip=args.norm_clip,
        momentum=args.momentum, weight_decay=args.weight_decay)

    compressor = gt.compressors[args.compression](args.density) \
        if args.compression != "none" else gt.compressors["none"]()
    writer = None
    if args.stats and rank == 0:
        from gtopkssgd_amd.utils.profiling import StatsWriter
        writer = StatsWriter(args.stats)
    opt = gt.DistributedOptimizer(
        trainer.optimizer, trainer.net.named_parameters(),
        compression=compressor, density=args.density,
        profile=args.profile or bool(writer), writer=writer,
        # clipping rescales grads after backward — incompatible with the
        # backward-overlapped EF accumulate
        overlap_sparse=args.norm_clip is None)

    start_epoch = 0
    if args.resume:
        from gtopkssgd_amd.checkpoint import latest_checkpoint, \
            load_checkpoint
        path = latest_checkpoint(args.checkpoint_dir, run_tag)
        if path:
            start_epoch = load_checkpoint(path, trainer.net, opt,
                                          map_location=str(trainer.device),
                                          rank=rank) + 1
            logger.info("resumed from %s (next epoch %d)", path,
                        start_epoch)
        else:
            logger.info("no checkpoint found for %s — fresh start", run_tag)

    iters_per_epoch = args.iters_per_epoch or max(
        1, len(trainer.trainloader) // args.nsteps_update)
    logger.info("run %s: world=%d iters/epoch=%d", run_tag, world,
                iters_per_epoch)

    prof = None
    if args.torch_profile and rank == 0:
        from torch.profiler import (ProfilerActivity, profile, schedule,
                                    tensorboard_trace_handler)
        prof = profile(
            activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
            schedule=schedule(wait=2, warmup=2, active=3, repeat=1),
            on_trace_ready=tensorboard_trace_handler(args.torch_profile))
        prof.__enter__()

    for epoch in range(start_epoch, args.max_epochs):
        lr = trainer.adjust_learning_rate(epoch, opt)
        t0 = time.time()
        for it in range(iters_per_epoch):
            loss = trainer.train(args.nsteps_update, dist_optimizer=opt)
            trainer.update_model(dist_optimizer=opt)
            if prof is not None:
                prof.step()
            if rank == 0 and (it + 1) % 50 == 0:
                wall = time.time() - t0
                seen = (it + 1) * args.nsteps_update * args.batch_size
                logger.info("epoch %d it %d/%d loss %.4f lr %.4g "
                            "Speed %.1f samples/s/GPU", epoch, it + 1,
                            iters_per_epoch, loss, lr, seen / wall)
        wall = time.time() - t0
        results = trainer.test(epoch)  # all ranks: metrics are all-reduced
        if rank == 0:
            sps = iters_per_epoch * args.nsteps_update * args.batch_size / wall
            logger.info("epoch %d done in %.1fs — %.1f samples/s/GPU (%s)",
                        epoch, wall, sps, trainer.timer.summary())
            if metrics:
                metrics.log(epoch=epoch, lr=lr, samples_per_sec=sps,
                            world=world, **results)
        # every rank: rank 0 writes model+optimizer, each rank its EF sidecar
        trainer.save(args.checkpoint_dir, run_tag, epoch, optimizer=opt)
        trainer.timer.reset()

    if prof is not None:
        prof.__exit__(None, None, None)
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
