"""Checkpoint/resume correctness, including per-rank EF residuals at
world > 1 (the EF residual is rank-local state and must
be restored rank-locally, never from rank 0's copy)."""
import os
import tempfile

import torch
import torch.distributed as dist
import torch.nn as nn

import gtopkssgd_amd as gt
from gtopkssgd_amd.checkpoint import (latest_checkpoint, load_checkpoint,
                                      save_checkpoint)
from utils_dist import run_in_procs


def _make_model(seed=11):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(32, 64), nn.ReLU(), nn.Linear(64, 4))


def _make_opt(model, density=0.05):
    inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9,
                            weight_decay=1e-4)
    return gt.DistributedOptimizer(
        inner, model.named_parameters(),
        compression=gt.compressors["gtopk"](density), density=density)


def _batch(rank, step):
    g = torch.Generator().manual_seed(10_000 * (rank + 1) + step)
    x = torch.randn(16, 32, generator=g)
    y = torch.randint(0, 4, (16,), generator=g)
    return x, y


def _steps(model, opt, rank, start, count):
    crit = nn.CrossEntropyLoss()
    for s in range(start, start + count):
        x, y = _batch(rank, s)
        opt.zero_grad()
        crit(model(x), y).backward()
        opt.step()


def _worker_resume_bitwise(rank, world, tmpdir):
    # Run A: 6 uninterrupted steps (per-rank distinct data => per-rank
    # distinct EF residuals, the case round 1 got wrong).
    model_a = _make_model()
    opt_a = _make_opt(model_a)
    _steps(model_a, opt_a, rank, 0, 6)
    params_a = opt_a.flat_param.clone()
    resid_a = opt_a.compression.residuals["flat"].clone()
    mom_a = opt_a.flat_momentum.clone()

    # Run B: 3 steps -> checkpoint -> fresh objects -> resume -> 3 steps.
    model_b = _make_model()
    opt_b = _make_opt(model_b)
    _steps(model_b, opt_b, rank, 0, 3)
    save_checkpoint(tmpdir, "rt", 0, model_b, opt_b, rank=rank)
    dist.barrier()  # rank 0's model file must exist before others read

    model_c = _make_model()
    opt_c = _make_opt(model_c)
    path = latest_checkpoint(tmpdir, "rt")
    assert path is not None
    load_checkpoint(path, model_c, opt_c, rank=rank)
    _steps(model_c, opt_c, rank, 3, 3)

    assert torch.equal(opt_c.flat_param, params_a), \
        f"rank {rank}: params differ after resume"
    assert torch.equal(opt_c.compression.residuals["flat"], resid_a), \
        f"rank {rank}: EF residual differs after resume"
    assert torch.equal(opt_c.flat_momentum, mom_a), \
        f"rank {rank}: momentum differs after resume"


def test_resume_bitwise_world2(tmp_path):
    run_in_procs(_worker_resume_bitwise, 2, args=(str(tmp_path),))


def _worker_sidecar_missing(rank, world, tmpdir):
    """Resume with a missing sidecar (world-size change): the rank must
    get a ZERO residual, not rank 0's."""
    model = _make_model()
    opt = _make_opt(model)
    _steps(model, opt, rank, 0, 2)
    if rank == 0:
        save_checkpoint(tmpdir, "rt", 0, model, opt, rank=0)
    dist.barrier()
    model2 = _make_model()
    opt2 = _make_opt(model2)
    _steps(model2, opt2, rank, 0, 1)  # dirty the residual
    path = latest_checkpoint(tmpdir, "rt")
    load_checkpoint(path, model2, opt2, rank=rank)
    resid = opt2.compression.residuals["flat"]
    if rank == 0:
        assert torch.equal(resid, opt.compression.residuals["flat"])
    else:
        assert torch.count_nonzero(resid) == 0, \
            "rank!=0 must not adopt rank 0's residual"


def test_resume_missing_sidecar_zeros_residual_world2(tmp_path):
    run_in_procs(_worker_sidecar_missing, 2, args=(str(tmp_path),))


def _worker_sidecar_files(rank, world, tmpdir):
    model = _make_model()
    opt = _make_opt(model)
    _steps(model, opt, rank, 0, 1)
    save_checkpoint(tmpdir, "rt", 3, model, opt, rank=rank)
    dist.barrier()
    if rank == 0:
        names = sorted(os.listdir(os.path.join(tmpdir, "rt")))
        assert names == ["ef-3-rank0.pth", "ef-3-rank1.pth",
                         "model-3.pth"], names


def test_sidecar_files_written_per_rank(tmp_path):
    run_in_procs(_worker_sidecar_files, 2, args=(str(tmp_path),))
