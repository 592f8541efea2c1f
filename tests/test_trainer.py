"""DLTrainer loop + checkpoint round-trip on CPU."""
import os

import torch

import gtopkssgd_amd as gt
from gtopkssgd_amd.checkpoint import (latest_checkpoint, load_checkpoint,
                                      save_checkpoint)
from gtopkssgd_amd.trainer import DLTrainer


def _tiny_trainer(dnn="mnistffn", dataset="mnist", **kw):
    return DLTrainer(dnn=dnn, dataset=dataset, batch_size=8,
                     device="cpu", amp=False, num_workers=0,
                     n_samples=64, **kw)


def test_trainer_steps_and_timers():
    tr = _tiny_trainer()
    opt = gt.DistributedOptimizer(tr.optimizer, tr.net.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01))
    loss0 = tr.train(2, dist_optimizer=opt)
    tr.update_model(dist_optimizer=opt)
    assert tr.train_iter == 2
    assert set(tr.timer.totals) >= {"io", "fwd", "bwd", "update"}
    # flat grad zeroed after update
    assert torch.count_nonzero(opt.flat_grad) == 0


def test_trainer_lr_schedule():
    tr = _tiny_trainer(dnn="resnet20", dataset="cifar10")
    assert tr.adjust_learning_rate(0) == tr.base_lr
    assert abs(tr.adjust_learning_rate(80) - tr.base_lr * 0.1) < 1e-12
    assert abs(tr.adjust_learning_rate(120) - tr.base_lr * 0.01) < 1e-12


def test_trainer_eval_metrics():
    tr = _tiny_trainer()
    m = tr.test(0)
    assert "top1" in m and 0.0 <= m["top1"] <= 1.0


def test_trainer_lstm_path():
    tr = DLTrainer(dnn="lstm", dataset="ptb", batch_size=4, device="cpu",
                   amp=False, num_workers=0, n_samples=16,
                   model_kwargs={"vocab_size": 100, "hidden": 16})
    loss = tr.train(1)
    assert loss > 0
    m = tr.test(0)
    assert "perplexity" in m


def test_checkpoint_roundtrip(tmp_path):
    tr = _tiny_trainer()
    opt = gt.DistributedOptimizer(tr.optimizer, tr.net.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.05))
    tr.train(1, dist_optimizer=opt)
    tr.update_model(dist_optimizer=opt)
    path = save_checkpoint(str(tmp_path), "run", 0, tr.net, opt)
    assert os.path.exists(path)
    assert latest_checkpoint(str(tmp_path), "run") == path

    tr2 = _tiny_trainer()
    opt2 = gt.DistributedOptimizer(tr2.optimizer, tr2.net.named_parameters(),
                                   compression=gt.compressors["gtopk"](0.05))
    epoch = load_checkpoint(path, tr2.net, opt2)
    assert epoch == 0
    for p1, p2 in zip(tr.net.parameters(), tr2.net.parameters()):
        assert torch.allclose(p1, p2)
    # EF residual state survives resume (documented extension)
    assert torch.allclose(opt2.compression.residuals["flat"],
                          opt.compression.residuals["flat"])
    assert torch.allclose(opt2.flat_momentum, opt.flat_momentum)


def test_trainer_an4_ctc_path():
    tr = DLTrainer(dnn="lstman4", dataset="an4", batch_size=4, device="cpu",
                   amp=False, num_workers=0, n_samples=16,
                   model_kwargs={"hidden": 16})
    opt = gt.DistributedOptimizer(tr.optimizer, tr.net.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01))
    loss = tr.train(1, dist_optimizer=opt)
    tr.update_model(dist_optimizer=opt)
    assert loss > 0
    m = tr.test(0)
    assert "ctc_loss" in m


def test_lr_schedule_all_datasets():
    for ds, dnn in [("mnist", "mnistffn"), ("ptb", "lstm")]:
        kw = {"model_kwargs": {"vocab_size": 50, "hidden": 16}} \
            if dnn == "lstm" else {}
        tr = DLTrainer(dnn=dnn, dataset=ds, batch_size=4, device="cpu",
                       amp=False, num_workers=0, n_samples=16, **kw)
        assert tr.adjust_learning_rate(0) == tr.base_lr


def test_latest_checkpoint_empty_dir(tmp_path):
    assert latest_checkpoint(str(tmp_path), "nothing") is None


def test_get_acc_resets():
    tr = _tiny_trainer()
    tr.train(1)
    a = tr.get_acc()
    assert a > 0
    assert tr._loss_count == 0
