"""Self-arming multi-GPU RCCL tests: when the box has
>= 2 visible devices, torchrun-spawn world=2 over the real nccl(=RCCL)
backend; on 1-GPU boxes they skip cleanly.  This makes any future
multi-GPU driver run of `pytest -m gpu` exercise the xGMI p2p merge,
bucketed dense all-reduce, and NCCL metric sync automatically.
"""
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(REPO, "tests", "workers_multigpu.py")


def _need_2gpus():
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 GPUs")


def _torchrun(check: str, port: int, nproc: int = 2, timeout: int = 420):
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
         "--master-port", str(port), WORKER, "--check", check],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-3000:])
    assert f"MULTIGPU OK {check}" in r.stdout


def test_gtopk_merge_rccl_world2():
    _need_2gpus()
    _torchrun("merge", 29531)


def test_resnet20_train_gtopk_rccl_world2():
    _need_2gpus()
    _torchrun("train", 29532)


def test_resnet20_train_dense_overlap_rccl_world2():
    _need_2gpus()
    _torchrun("dense", 29533)
