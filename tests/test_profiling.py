"""Phase timer + stats writer (CPU: timer disabled path; writer JSONL)."""
import json

import torch

import gtopkssgd_amd as gt
from gtopkssgd_amd.models import MnistFFN
from gtopkssgd_amd.utils.profiling import GpuPhaseTimer, StatsWriter


def test_timer_disabled_on_cpu_is_noop():
    t = GpuPhaseTimer(enabled=True)  # no CUDA -> disabled
    t.start("x")
    t.stop("x")
    assert t.collect() == {}
    assert t.summary() == ""


def test_stats_writer_jsonl(tmp_path):
    path = str(tmp_path / "stats.jsonl")
    w = StatsWriter(path)
    w.add_scalar("a", 1.5, 3)
    w.add_scalar("b", 2.0, 4)
    rows = [json.loads(l) for l in open(path)]
    assert rows[0]["tag"] == "a" and rows[0]["value"] == 1.5


def test_optimizer_with_profile_and_writer(tmp_path):
    torch.manual_seed(0)
    model = MnistFFN(hidden=16)
    inner = torch.optim.SGD(model.parameters(), lr=0.1)
    w = StatsWriter(str(tmp_path / "opt.jsonl"))
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.05),
                                  profile=True, writer=w)
    x = torch.randn(4, 1, 28, 28)
    y = torch.randint(0, 10, (4,))
    loss = torch.nn.CrossEntropyLoss()(model(x), y)
    loss.backward()
    opt.step()
    assert opt.train_iter == 1
