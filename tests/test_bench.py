"""bench.py contract: the driver launches it standalone (N=1) and via
torchrun (N>1).  GTOPK_BENCH_CPU=1 runs the identical code path on
CPU/gloo so the multi-rank plumbing (env parsing, db-prime barrier,
MAX-reduce, one-line JSON contract) is exercised here — the real GPU
numbers come from gpurun / the driver."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def _parse_json_line(stdout: str) -> dict:
    for line in stdout.splitlines():
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


def _common_checks(out: dict, n: int):
    assert out["n_gpus"] == n
    assert out["value"] > 0
    assert out["ms_per_step"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["dtype"] == "bf16"
    assert out["data"] == "synthetic"
    assert out["config"]["model"] == "resnet20"


def test_bench_single_process_cpu():
    env = dict(os.environ, GTOPK_BENCH_CPU="1")
    r = subprocess.run(
        [sys.executable, BENCH, "--model", "resnet20", "--batch-size",
         "8", "--steps", "2", "--warmup", "1"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    out = _parse_json_line(r.stdout)
    _common_checks(out, 1)
    assert out["steps"] == 2 and out["warmup"] == 1


def test_bench_torchrun_world2_cpu():
    """The exact launch shape the driver uses for SCALE runs."""
    env = dict(os.environ, GTOPK_BENCH_CPU="1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", BENCH, "--gpus", "2", "--model",
         "resnet20", "--batch-size", "8", "--steps", "2", "--warmup",
         "1"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    out = _parse_json_line(r.stdout)
    _common_checks(out, 2)
    # whole-job aggregate: 2 ranks x 8 images / step
    assert out["config"]["global_batch"] == 16
