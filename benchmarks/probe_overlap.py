#!/usr/bin/env python3
"""Bisect the world=1 cost of backward-overlapped compression.

Fresh-box A/B (profiles/RESULTS.md): overlap ON costs ~1.2 ms/step on
ResNet-50 bs512.  Candidate costs: (a) per-param Python hook calls,
(b) per-bucket event create/record/wait + stream switches, (c) the
side-stream kernels contending with backward for HBM/L2.

Measured verdict (2026-09-13): (b) — the side stream + events — was the
whole 1.2 ms; hooks and kernels were free.  The shipped implementation
now launches the bucket kernels on the current (backward) stream, which
matched overlap-off throughput; the "sidestream" variant here
reproduces the abandoned design for the record.

Variants:
  off        overlap disabled (sync select in step)
  noop       hooks fire but buckets do nothing (isolates (a))
  full       the shipped in-stream hook overlap
  sidestream the abandoned side-stream + events design
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import torch  # noqa: E402
import torch.nn as nn  # noqa: E402

import gtopkssgd_amd as gt  # noqa: E402
from gtopkssgd_amd import models as zoo, ops  # noqa: E402


def prime_db():
    import tarfile
    tar = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "profiles", "miopen_udb.tar.gz")
    if os.path.exists(tar):
        with tarfile.open(tar) as t:
            t.extractall(os.path.expanduser("~"))


def run(variant, steps=30, warmup=10, batch=512):
    torch.manual_seed(0)
    torch.backends.cudnn.benchmark = False
    device = torch.device("cuda", 0)
    model = zoo.create_net("resnet50").to(device) \
        .to(memory_format=torch.channels_last)
    inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                            weight_decay=1e-4)
    opt = gt.DistributedOptimizer(
        inner, model.named_parameters(),
        compression=gt.compressors["gtopk"](0.001),
        overlap_sparse=variant != "off")
    if variant == "noop":
        opt._issue_sparse_bucket = lambda b: None
    elif variant == "sidestream":
        stream = torch.cuda.Stream()
        state = {"ev": None}

        def issue(b):
            ev = torch.cuda.Event()
            ev.record()
            stream.wait_event(ev)
            with torch.cuda.stream(stream):
                if opt._n_issued == 0:
                    ops.ef_hist_zero(opt.device)
                ops.ef_accum_hist_range(opt.compression.residuals["flat"],
                                        opt.flat_grad, b["start"], b["end"])
            b["issued"] = True
            opt._n_issued += 1
            if opt._n_issued == len(opt._buckets):
                with torch.cuda.stream(stream):
                    opt._overlap_select = opt.compression.compress_finish(
                        "flat", out_val=opt._sel_val, out_idx=opt._sel_idx)
                done = torch.cuda.Event()
                done.record(stream)
                state["ev"] = done
        opt._issue_sparse_bucket = issue
        orig_finish = opt._finish_sparse_overlap

        def finish():
            if state["ev"] is not None:
                torch.cuda.current_stream().wait_event(state["ev"])
                state["ev"] = None
            return orig_finish()
        opt._finish_sparse_overlap = finish
    crit = nn.CrossEntropyLoss()
    gen = torch.Generator().manual_seed(1)
    xs = [torch.randn(batch, 3, 224, 224, generator=gen)
          .to(device, memory_format=torch.channels_last) for _ in range(2)]
    ys = [torch.randint(0, 1000, (batch,), generator=gen).to(device)
          for _ in range(2)]

    def step(i):
        opt.zero_grad()
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss = crit(model(xs[i % 2]), ys[i % 2])
        loss.backward()
        opt.step()

    model.train()
    for i in range(warmup):
        step(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step(i)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / steps * 1000
    print(json.dumps({"variant": variant, "ms_per_step": round(ms, 2),
                      "img_per_sec": round(batch / ms * 1000, 1)}),
          flush=True)
    del model, inner, opt, xs, ys
    torch.cuda.empty_cache()


if __name__ == "__main__":
    prime_db()
    order = ["off", "full", "noop", "sidestream", "off", "full"]
    for v in order:
        run(v)
