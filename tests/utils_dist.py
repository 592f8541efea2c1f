"""Helpers for multi-process CPU (gloo) tests."""
import os
import tempfile

import torch.distributed as dist
import torch.multiprocessing as mp


def run_in_procs(fn, world_size=2, args=(), timeout=120):
    """Spawn world_size processes, init gloo via a file store, run
    fn(rank, world_size, *args) in each, propagate failures."""
    with tempfile.TemporaryDirectory() as d:
        init_file = os.path.join(d, "init")
        ctx = mp.get_context("spawn")
        procs = []
        for r in range(world_size):
            p = ctx.Process(target=_entry,
                            args=(fn, r, world_size, init_file, args))
            p.start()
            procs.append(p)
        failed = []
        for r, p in enumerate(procs):
            p.join(timeout)
            if p.is_alive():
                p.terminate()
                p.join(5)
                failed.append((r, "timeout"))
            elif p.exitcode != 0:
                failed.append((r, f"exit={p.exitcode}"))
        assert not failed, f"worker failures: {failed}"


def _entry(fn, rank, world_size, init_file, args):
    dist.init_process_group(
        backend="gloo", init_method=f"file://{init_file}",
        rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()
