#!/usr/bin/env python3
"""Convergence comparison: gTop-k vs Top-k vs dense S-SGD.

Reference parity: the reference validates gTop-k by
comparing loss/accuracy curves against dense SGD; its paper reports a
≲1% accuracy gap at ρ=0.001.  With no network for real datasets this
harness uses the learnable synthetic task (class-mean Gaussians).

    python benchmarks/convergence.py --dnn resnet20 --epochs 5 \
        --iters-per-epoch 50 --density 0.001 --out curves.jsonl

Runs every mode with identical seeds/batches and prints a comparison
table; writes per-epoch JSONL.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn as nn  # noqa: E402

import gtopkssgd_amd as gt  # noqa: E402
from gtopkssgd_amd import models as zoo  # noqa: E402
from gtopkssgd_amd.data.synthetic import SyntheticImages  # noqa: E402


def run_mode(mode, args, device):
    torch.manual_seed(args.seed)
    net = zoo.create_net(args.dnn, "cifar10").to(device)
    if device.type == "cuda":
        net = net.to(memory_format=torch.channels_last)
    inner = torch.optim.SGD(net.parameters(), lr=args.lr, momentum=0.9,
                            weight_decay=5e-4)
    comp = (gt.compressors["none"]() if mode == "dense"
            else gt.compressors[mode](args.density))
    opt = gt.DistributedOptimizer(inner, net.named_parameters(),
                                  compression=comp, density=args.density,
                                  mode=mode)
    crit = nn.CrossEntropyLoss()
    train = SyntheticImages("cifar10", args.iters_per_epoch * args.batch_size,
                            seed=args.seed, learnable=True)
    test = SyntheticImages("cifar10", 2048, seed=args.seed + 1,
                           learnable=True)
    test.class_means = train.class_means  # same task
    history = []
    amp = device.type == "cuda"
    for epoch in range(args.epochs):
        net.train()
        total = 0.0
        perm = torch.randperm(len(train),
                              generator=torch.Generator().manual_seed(epoch))
        for it in range(args.iters_per_epoch):
            sel = perm[it * args.batch_size:(it + 1) * args.batch_size]
            xs = torch.stack([train[int(j)][0] for j in sel]).to(device)
            ys = torch.tensor([train[int(j)][1] for j in sel]).to(device)
            if device.type == "cuda":
                xs = xs.contiguous(memory_format=torch.channels_last)
            opt.zero_grad()
            with torch.autocast("cuda", torch.bfloat16, enabled=amp):
                loss = crit(net(xs), ys)
            loss.backward()
            opt.step()
            total += loss.item()
        net.eval()
        correct = 0
        with torch.no_grad():
            for j0 in range(0, len(test), 256):
                xs = torch.stack([test[j][0]
                                  for j in range(j0, min(j0 + 256,
                                                         len(test)))])
                ys = torch.tensor([test[j][1]
                                   for j in range(j0, min(j0 + 256,
                                                          len(test)))])
                out = net(xs.to(device))
                correct += int((out.argmax(1).cpu() == ys).sum())
        acc = correct / len(test)
        history.append({"mode": mode, "epoch": epoch,
                        "train_loss": total / args.iters_per_epoch,
                        "test_acc": acc})
        print(f"[{mode}] epoch {epoch}: loss "
              f"{total / args.iters_per_epoch:.4f} acc {acc:.4f}",
              flush=True)
    return history


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--dnn", default="resnet20")
    p.add_argument("--modes", default="dense,gtopk,topk")
    p.add_argument("--epochs", type=int, default=5)
    p.add_argument("--iters-per-epoch", type=int, default=50)
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--density", type=float, default=0.001)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--seed", type=int, default=7)
    p.add_argument("--out", default=None)
    args = p.parse_args()
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    all_hist = []
    final = {}
    for mode in args.modes.split(","):
        t0 = time.time()
        hist = run_mode(mode, args, device)
        all_hist += hist
        final[mode] = hist[-1]
        print(f"[{mode}] done in {time.time() - t0:.1f}s")
    if args.out:
        with open(args.out, "w") as f:
            for row in all_hist:
                f.write(json.dumps(row) + "\n")
    print("\nfinal:", json.dumps(final))
    if "dense" in final:
        for mode, row in final.items():
            if mode != "dense":
                gap = final["dense"]["test_acc"] - row["test_acc"]
                print(f"accuracy gap {mode} vs dense: {gap:+.4f}")


if __name__ == "__main__":
    main()
