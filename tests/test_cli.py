"""train.py CLI end-to-end (single process, CPU, tiny)."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_train_cli_single_process(tmp_path):
    metrics = str(tmp_path / "m.jsonl")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--dnn", "mnistffn", "--dataset", "mnist", "--batch-size", "8",
         "--max-epochs", "1", "--iters-per-epoch", "3", "--n-samples", "64",
         "--density", "0.05", "--device", "cpu", "--no-amp",
         "--num-workers", "0", "--metrics", metrics,
         "--checkpoint-dir", str(tmp_path / "ckpt")],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    rows = [json.loads(l) for l in open(metrics)]
    assert rows and "samples_per_sec" in rows[0]
    run_dirs = os.listdir(str(tmp_path / "ckpt"))
    assert len(run_dirs) == 1
    assert any(f.startswith("model-") for f in
               os.listdir(os.path.join(str(tmp_path / "ckpt"), run_dirs[0])))


def test_train_cli_gloo_world2(tmp_path):
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", os.path.join(REPO, "train.py"),
         "--dnn", "mnistffn", "--dataset", "mnist", "--batch-size", "8",
         "--max-epochs", "1", "--iters-per-epoch", "2", "--n-samples", "64",
         "--density", "0.05", "--device", "cpu", "--no-amp",
         "--num-workers", "0",
         "--checkpoint-dir", str(tmp_path / "ckpt")],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])


def _torchrun_train(nproc: int, port: int, tmp_path, timeout=420):
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "train.py"),
         "--dnn", "mnistffn", "--dataset", "mnist", "--batch-size", "8",
         "--max-epochs", "1", "--iters-per-epoch", "2", "--n-samples", "64",
         "--density", "0.05", "--device", "cpu", "--no-amp",
         "--num-workers", "0",
         "--checkpoint-dir", str(tmp_path / "ckpt")],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])


def test_train_cli_gloo_world3(tmp_path):
    """Non-power-of-2 world: the halving merge's peer-existence edge
    (rank 2 pairs with a missing rank 3) end-to-end through torchrun."""
    _torchrun_train(3, 29518, tmp_path)


def test_train_cli_gloo_world4(tmp_path):
    """Two full merge rounds (2/4/8-proc single-node)."""
    _torchrun_train(4, 29519, tmp_path)


def test_collective_self_benchmark_cpu():
    r = subprocess.run(
        [sys.executable, "-m", "gtopkssgd_amd.parallel.allreducer",
         "--n", "10000", "--density", "0.01", "--iters", "3",
         "--warmup", "1", "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-1000:]
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert out["collective"] == "gtopk" and out["k"] == 100


def test_train_cli_resume(tmp_path):
    base = [sys.executable, os.path.join(REPO, "train.py"),
            "--dnn", "mnistffn", "--dataset", "mnist", "--batch-size", "8",
            "--iters-per-epoch", "2", "--n-samples", "32",
            "--density", "0.05", "--device", "cpu", "--no-amp",
            "--num-workers", "0", "--run-tag", "rt",
            "--checkpoint-dir", str(tmp_path / "ckpt")]
    r1 = subprocess.run(base + ["--max-epochs", "1"], cwd=REPO,
                        capture_output=True, text=True, timeout=300)
    assert r1.returncode == 0, r1.stderr[-1000:]
    r2 = subprocess.run(base + ["--max-epochs", "3", "--resume"], cwd=REPO,
                        capture_output=True, text=True, timeout=300)
    assert r2.returncode == 0, r2.stderr[-1000:]
    assert "resumed from" in r2.stderr + r2.stdout
    names = os.listdir(str(tmp_path / "ckpt" / "rt"))
    assert "model-2.pth" in names
