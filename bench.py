#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50, ImageNet-shape synthetic, gTop-k
density=0.001, bf16 autocast — the BASELINE.json headline metric
(images/sec whole job over N MI355X GPUs, weak scaling).

Single GPU:   python bench.py --gpus 1 --steps 30 --warmup 10
Multi GPU:    python -m torch.distributed.run --nnodes=1 \
                 --nproc-per-node N --master-addr 127.0.0.1 bench.py \
                 --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""
import argparse
import json
import os
import time

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402
import torch.nn as nn  # noqa: E402

import gtopkssgd_amd as gt  # noqa: E402
from gtopkssgd_amd.models import resnet50  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=256,
                   help="per-GPU batch size (weak scaling)")
    p.add_argument("--density", type=float, default=0.001)
    p.add_argument("--mode", default="gtopk",
                   choices=["gtopk", "topk", "dense"])
    p.add_argument("--model", default="resnet50")
    p.add_argument("--lr", type=float, default=0.1)
    return p.parse_args()


def _prime_miopen_db():
    """Extract a committed MIOpen find-db/binary-cache snapshot (captured
    on an MI355X box) into $HOME so conv algo selection skips the
    multi-minute exhaustive find on fresh machines.  A cache of MIOpen's
    own tuning results — not of benchmark outputs."""
    import tarfile
    tar = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "profiles", "miopen_udb.tar.gz")
    home = os.path.expanduser("~")
    if os.path.exists(tar) and not os.environ.get("GTOPK_NO_DB_PRIME"):
        try:
            with tarfile.open(tar) as t:
                t.extractall(home)
        except Exception as exc:  # pragma: no cover
            print(f"# miopen db prime failed: {exc}", flush=True)


def main():
    args = parse_args()
    assert torch.cuda.is_available(), "bench.py requires an MI355X GPU"
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        dist.init_process_group(backend="nccl")
        torch.cuda.set_device(local_rank)
    # rank 0 extracts the MIOpen cache snapshot; others wait (a
    # concurrent multi-rank extract would race on the same files)
    if local_rank == 0:
        _prime_miopen_db()
    if world > 1:
        dist.barrier()
    device = torch.device("cuda", local_rank)
    torch.manual_seed(1234)
    # MIOpen: exhaustive-find conv algo selection (amortised in warmup);
    # without it MIOpen's FAST fallback picks slow zero-workspace
    # solvers.  GTOPK_CUDNN_BENCHMARK=0 disables for A/B testing.
    torch.backends.cudnn.benchmark = \
        os.environ.get("GTOPK_CUDNN_BENCHMARK", "1") == "1"

    model = resnet50().to(device).to(memory_format=torch.channels_last)
    inner = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                            weight_decay=1e-4)
    comp = (gt.compressors["none"]() if args.mode == "dense"
            else gt.compressors[args.mode](args.density))
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=comp, density=args.density)
    crit = nn.CrossEntropyLoss()

    B = args.batch_size
    n_buf = 4  # pre-staged synthetic batches, cycled
    gen = torch.Generator(device="cpu").manual_seed(4321 + rank)
    xs = [torch.randn(B, 3, 224, 224, generator=gen)
          .to(device, memory_format=torch.channels_last) for _ in range(n_buf)]
    ys = [torch.randint(0, 1000, (B,), generator=gen).to(device)
          for _ in range(n_buf)]

    def step(i):
        x, y = xs[i % n_buf], ys[i % n_buf]
        opt.zero_grad()
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss = crit(model(x), y)
        loss.backward()
        opt.step()
        return loss

    model.train()
    tw = time.perf_counter()
    for i in range(args.warmup):
        step(i)
    torch.cuda.synchronize()
    if rank == 0:
        print(f"# warmup: {time.perf_counter() - tw:.1f}s "
              f"({args.warmup} steps)", flush=True)
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        value = world * B * args.steps / elapsed
        out = {
            "metric": "images/sec (whole node) ResNet-50 gTop-k "
                      "density=0.001",
            "value": round(value, 1),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": world * B,
                "batch_per_gpu": B,
                "density": args.density,
                "parallelism": f"dp{world}",
                "compression": args.mode,
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
