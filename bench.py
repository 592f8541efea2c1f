#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50, ImageNet-shape synthetic, gTop-k
density=0.001, bf16 autocast — the BASELINE.json headline metric
(images/sec whole job over N MI355X GPUs, weak scaling).

Also runs the supporting BASELINE configs via --model
(resnet20/vgg16 CIFAR-shape, lstm PTB-shape, alexnet).

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
from gtopkssgd_amd import models as zoo  # noqa: E402

# model -> (builder kwargs dataset-shape, default per-GPU batch, unit)
_CONFIGS = {
    # bs768 measured 9200 img/s vs bs512's 8861 under the same FAST
    # find mode + committed MIOpen db (0.8 s warmup) — RESULTS.md sweep
    "resnet50": dict(shape=(3, 224, 224), classes=1000, batch=768),
    "alexnet": dict(shape=(3, 224, 224), classes=1000, batch=512),
    "resnet20": dict(shape=(3, 32, 32), classes=10, batch=2048),
    "vgg16": dict(shape=(3, 32, 32), classes=10, batch=1024),
    "lstm": dict(seq_len=35, vocab=10000, batch=512),
    "lstman4": dict(freq=161, time=200, labels=29, tmax=20, batch=64),
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=0,
                   help="per-GPU batch size (0 = model default)")
    p.add_argument("--density", type=float, default=0.001)
    p.add_argument("--mode", default="gtopk",
                   choices=["gtopk", "topk", "dense"])
    p.add_argument("--model", default="resnet50",
                   choices=sorted(_CONFIGS))
    p.add_argument("--no-channels-last", action="store_true")
    p.add_argument("--lr", type=float, default=0.1)
    return p.parse_args()


def _prime_miopen_db() -> bool:
    """Extract a committed MIOpen find-db/perf-db/binary-cache snapshot
    (captured on an MI355X box after exhaustive find over every bench
    config) into $HOME.  A cache of MIOpen's own tuning results — not of
    benchmark outputs.  Returns True when primed."""
    import tarfile
    tar = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "profiles", "miopen_udb.tar.gz")
    home = os.path.expanduser("~")
    if os.path.exists(tar) and not os.environ.get("GTOPK_NO_DB_PRIME"):
        try:
            with tarfile.open(tar) as t:
                t.extractall(home)
            return True
        except Exception as exc:  # pragma: no cover
            print(f"# miopen db prime failed: {exc}", flush=True)
    return False


def main():
    args = parse_args()
    # GTOPK_BENCH_CPU=1: run the identical code path on CPU over gloo —
    # a plumbing test of the multi-rank bench (env parsing, barriers,
    # MAX-reduce, JSON contract), NOT a performance mode.  The driver's
    # GPU runs never set it.
    cpu_mode = os.environ.get("GTOPK_BENCH_CPU") == "1"
    if not cpu_mode:
        assert torch.cuda.is_available(), "bench.py requires an MI355X GPU"
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        dist.init_process_group(backend="gloo" if cpu_mode else "nccl")
        if not cpu_mode:
            torch.cuda.set_device(local_rank)
    # rank 0 extracts the MIOpen cache snapshot; others wait (a
    # concurrent multi-rank extract would race on the same files)
    primed = False
    if local_rank == 0 and not cpu_mode:
        primed = _prime_miopen_db()
    if world > 1:
        dist.barrier()
        primed = os.path.isdir(os.path.expanduser("~/.config/miopen"))
        # each rank gets its own user find-db copy: 8 ranks sharing one
        # db file serialize on MIOpen's file locks during find
        import shutil
        src_db = os.path.expanduser("~/.config/miopen")
        if os.path.isdir(src_db):
            rank_db = os.path.expanduser(f"~/.config/miopen_r{local_rank}")
            try:
                shutil.copytree(src_db, rank_db, dirs_exist_ok=True)
                os.environ["MIOPEN_USER_DB_PATH"] = rank_db
            except OSError as exc:
                print(f"# per-rank miopen db copy failed: {exc}",
                      flush=True)
    device = torch.device("cpu") if cpu_mode \
        else torch.device("cuda", local_rank)
    torch.manual_seed(1234)

    cfg = _CONFIGS[args.model]
    B = args.batch_size or cfg["batch"]
    # MIOpen solver selection.  Measured on MI355X (profiles/RESULTS.md):
    # benchmark=True re-runs the ~115 s exhaustive find in EVERY process
    # (neither the user find-db nor a same-box second process skips it),
    # while FAST find mode gives ResNet-50 throughput parity on a FRESH
    # box (8675.7 img/s, 0.7 s warmup vs 8656.0, 115 s) — its ImageNet
    # shapes are covered by MIOpen's system databases.  The CIFAR-shape
    # models and AlexNet are NOT covered (FAST fresh-box: resnet20 1850
    # vs 276547 img/s), so they keep the exhaustive find (their finds
    # are 36-50 s).  GTOPK_CUDNN_BENCHMARK=0/1 forces either path.
    force = os.environ.get("GTOPK_CUDNN_BENCHMARK")
    if force is not None:
        torch.backends.cudnn.benchmark = force == "1"
    elif args.model == "resnet50" and B in (256, 512, 768):
        os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
        torch.backends.cudnn.benchmark = False
    else:
        torch.backends.cudnn.benchmark = True
    is_lm = args.model == "lstm"
    is_ctc = args.model == "lstman4"
    channels_last = not (is_lm or is_ctc) and not args.no_channels_last
    if is_lm:
        model = zoo.lstm_ptb(cfg["vocab"], 1500).to(device)
    else:
        model = zoo.create_net(args.model).to(device)
        if channels_last:
            model = model.to(memory_format=torch.channels_last)
    inner = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                            weight_decay=1e-4)
    comp = (gt.compressors["none"]() if args.mode == "dense"
            else gt.compressors[args.mode](args.density))
    opt = gt.DistributedOptimizer(
        inner, model.named_parameters(), compression=comp,
        density=args.density,
        # GTOPK_OVERLAP=0 disables the backward-overlapped compression
        # (A/B knob; default on)
        overlap_sparse=os.environ.get("GTOPK_OVERLAP", "1") == "1")
    crit = nn.CTCLoss(blank=0, zero_infinity=True) if is_ctc \
        else nn.CrossEntropyLoss()

    n_buf = 4  # pre-staged synthetic batches, cycled
    gen = torch.Generator(device="cpu").manual_seed(4321 + rank)
    if is_lm:
        xs = [torch.randint(0, cfg["vocab"], (cfg["seq_len"] + 1, B),
                            generator=gen).to(device) for _ in range(n_buf)]
        hidden0 = model.init_hidden(B)
    elif is_ctc:
        xs = [torch.randn(B, 1, cfg["freq"], cfg["time"], generator=gen)
              .to(device) for _ in range(n_buf)]
        tlen = cfg["tmax"]
        ys = [torch.randint(1, cfg["labels"], (B * tlen,), generator=gen)
              .to(device) for _ in range(n_buf)]
        target_lengths = torch.full((B,), tlen, dtype=torch.long)
    else:
        xs = [torch.randn(B, *cfg["shape"], generator=gen)
              .to(device, memory_format=(torch.channels_last if channels_last
                                         else torch.contiguous_format))
              for _ in range(n_buf)]
        ys = [torch.randint(0, cfg["classes"], (B,), generator=gen).to(device)
              for _ in range(n_buf)]

    def step(i):
        opt.zero_grad()
        with torch.autocast(device_type=device.type,
                            dtype=torch.bfloat16, enabled=not cpu_mode):
            if is_lm:
                toks = xs[i % n_buf]
                logits, _ = model(toks[:-1], hidden0)
                loss = crit(logits.reshape(-1, logits.size(-1)),
                            toks[1:].reshape(-1))
            elif is_ctc:
                logits = model(xs[i % n_buf])  # [T, B, L]
                log_probs = logits.log_softmax(2).float()
                t = logits.size(0)
                input_lengths = torch.full((B,), t, dtype=torch.long)
                loss = crit(log_probs, ys[i % n_buf].view(B, -1),
                            input_lengths, target_lengths)
            else:
                loss = crit(model(xs[i % n_buf]), ys[i % n_buf])
        loss.backward()
        opt.step()
        return loss

    def _sync():
        if not cpu_mode:
            torch.cuda.synchronize()

    model.train()
    tw = time.perf_counter()
    for i in range(args.warmup):
        step(i)
    _sync()
    if rank == 0:
        print(f"# warmup: {time.perf_counter() - tw:.1f}s "
              f"({args.warmup} steps)", flush=True)
    if world > 1:
        dist.barrier()
    _sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    if world > 1:
        dist.barrier()
    _sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        per_step_items = B * (cfg["seq_len"] if is_lm else 1)
        value = world * per_step_items * args.steps / elapsed
        unit = ("tokens/sec" if is_lm else
                "utterances/sec" if is_ctc else "images/sec")
        metric = (f"{unit} (whole node) {args.model} {args.mode} "
                  f"density={args.density}")
        if args.model == "resnet50" and args.mode == "gtopk" \
                and args.density == 0.001:
            metric = ("images/sec (whole node) ResNet-50 gTop-k "
                      "density=0.001")
        out = {
            "metric": metric,
            "value": round(value, 1),
            "unit": unit,
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
