"""DistributedOptimizer end-to-end on CPU/gloo (BASELINE config 1 shape:
2-layer FFN, synthetic MNIST, world_size 2)."""
import torch
import torch.nn as nn

import gtopkssgd_amd as gt
from gtopkssgd_amd.models import MnistFFN
from utils_dist import run_in_procs


def _make_model(seed=7):
    torch.manual_seed(seed)
    return MnistFFN(hidden=32)


def _batch(seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(16, 1, 28, 28, generator=g)
    y = torch.randint(0, 10, (16,), generator=g)
    return x, y


def _train_reference(steps, lr=0.1, momentum=0.9, wd=1e-4):
    """Single-process SGD on the same (shared) batches = the expected
    result of 2-rank data-parallel training with identical batches."""
    model = _make_model()
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=momentum,
                          weight_decay=wd)
    crit = nn.CrossEntropyLoss()
    for s in range(steps):
        x, y = _batch(1000 + s)
        opt.zero_grad()
        crit(model(x), y).backward()
        opt.step()
    return torch.cat([p.data.view(-1) for p in model.parameters()])


def _worker_parity(rank, world, mode):
    """Same batch on every rank + rho=1 => must equal single-proc SGD."""
    model = _make_model()
    inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                            weight_decay=1e-4)
    comp = (gt.compressors["none"]() if mode == "dense"
            else gt.compressors[mode](1.0))
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=comp, density=1.0, mode=mode)
    crit = nn.CrossEntropyLoss()
    steps = 4
    for s in range(steps):
        x, y = _batch(1000 + s)
        opt.zero_grad()
        crit(model(x), y).backward()
        opt.step()
    flat = torch.cat([p.data.view(-1) for p in model.parameters()])
    expect = _train_reference(steps)
    assert torch.allclose(flat, expect, atol=1e-5), \
        (flat - expect).abs().max().item()


def test_dense_mode_parity_world2():
    run_in_procs(_worker_parity, 2, args=("dense",))


def test_gtopk_density1_equals_dense_world2():
    run_in_procs(_worker_parity, 2, args=("gtopk",))


def test_topk_density1_equals_dense_world2():
    run_in_procs(_worker_parity, 2, args=("topk",))


def test_single_process_gtopk_density1_parity():
    model = _make_model()
    inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                            weight_decay=1e-4)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](1.0))
    crit = nn.CrossEntropyLoss()
    for s in range(4):
        x, y = _batch(1000 + s)
        opt.zero_grad()
        crit(model(x), y).backward()
        opt.step()
    flat = torch.cat([p.data.view(-1) for p in model.parameters()])
    expect = _train_reference(4)
    assert torch.allclose(flat, expect, atol=1e-5)


def _worker_sparse_converges(rank, world):
    """rho=0.01 gTop-k on a learnable toy problem: loss must drop."""
    torch.manual_seed(7)
    model = nn.Sequential(nn.Linear(20, 64), nn.ReLU(), nn.Linear(64, 2))
    inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01))
    crit = nn.CrossEntropyLoss()
    g = torch.Generator().manual_seed(50 + rank)
    w_true = torch.randn(20, generator=g)
    losses = []
    for s in range(60):
        x = torch.randn(64, 20, generator=g)
        y = (x @ w_true > 0).long()
        opt.zero_grad()
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    first = sum(losses[:10]) / 10
    last = sum(losses[-10:]) / 10
    assert last < first * 0.7, (first, last)


def test_gtopk_sparse_converges_world2():
    run_in_procs(_worker_sparse_converges, 2)


def test_fused_vs_generic_update_paths_agree():
    """The fused flat-SGD path must match densify-into-grad + inner
    torch SGD step (single process, rho=0.05)."""
    results = []
    for force_generic in (False, True):
        model = _make_model()
        inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                                weight_decay=1e-4)
        opt = gt.DistributedOptimizer(
            inner, model.named_parameters(),
            compression=gt.compressors["gtopk"](0.05))
        if force_generic:
            assert opt._fused
            opt._fused = False  # keep flat params, use inner step
        crit = nn.CrossEntropyLoss()
        for s in range(5):
            x, y = _batch(2000 + s)
            opt.zero_grad()
            crit(model(x), y).backward()
            opt.step()
        results.append(torch.cat([p.data.view(-1)
                                  for p in model.parameters()]))
    assert torch.allclose(results[0], results[1], atol=1e-6), \
        (results[0] - results[1]).abs().max().item()


def test_warmup_dense_iters():
    model = _make_model()
    inner = torch.optim.SGD(model.parameters(), lr=0.1)
    opt = gt.DistributedOptimizer(inner,
                                  compression=gt.compressors["gtopk"](0.01),
                                  warmup_iters=2)
    # first two steps take the dense path, after that sparse
    p = inner.param_groups[0]["params"][0]
    for s in range(3):
        opt.zero_grad()
        loss = (p ** 2).sum()
        loss.backward()
        opt.step()
    assert opt.train_iter == 3


def test_channels_last_params_preserved():
    """Flat views must preserve channels_last conv-weight layout
    (MIOpen wants channels_last; .view would have failed/broken it)."""
    import torch.nn.functional as F
    torch.manual_seed(0)
    net = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                        nn.Conv2d(8, 4, 3, padding=1))
    net = net.to(memory_format=torch.channels_last)
    w0 = net[0].weight.detach().clone()
    inner = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9)
    opt = gt.DistributedOptimizer(inner, net.named_parameters(),
                                  compression=gt.compressors["gtopk"](1.0))
    assert net[0].weight.is_contiguous(memory_format=torch.channels_last)
    assert torch.equal(net[0].weight, w0)
    x = torch.randn(2, 3, 16, 16).contiguous(
        memory_format=torch.channels_last)
    loss = net(x).square().mean()
    loss.backward()
    assert net[0].weight.grad.is_contiguous(
        memory_format=torch.channels_last)
    opt.step()
    assert not torch.equal(net[0].weight, w0)  # update applied


def _worker_dense_overlap_parity(rank, world):
    """Bucketed overlap path must produce the same result as the
    one-shot dense all-reduce (rho irrelevant: mode=dense)."""
    import gtopkssgd_amd.parallel.optimizer as om
    om._DistributedOptimizer.BUCKET_ELEMS = 4096  # several buckets
    results = {}
    for overlap in (True, False):
        model = _make_model()
        inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
        opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                      compression=gt.compressors["none"](),
                                      mode="dense", overlap_dense=overlap)
        if overlap:
            # tiny buckets so several fire per backward
            assert opt._buckets
        crit = nn.CrossEntropyLoss()
        for s in range(3):
            x, y = _batch(3000 + s + rank)
            opt.zero_grad()
            crit(model(x), y).backward()
            opt.step()
        results[overlap] = torch.cat([p.data.view(-1)
                                      for p in model.parameters()])
    assert torch.allclose(results[True], results[False], atol=1e-6), \
        (results[True] - results[False]).abs().max().item()


def test_dense_overlap_parity_world2():
    run_in_procs(_worker_dense_overlap_parity, 2)


def _worker_overlap_with_accumulation(rank, world):
    """set_accumulation_boundary must keep partial gradients local."""
    import gtopkssgd_amd.parallel.optimizer as om
    om._DistributedOptimizer.BUCKET_ELEMS = 4096
    model = _make_model()
    inner = torch.optim.SGD(model.parameters(), lr=0.1)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["none"](),
                                  mode="dense")
    crit = nn.CrossEntropyLoss()
    for s in range(2):
        opt.zero_grad()
        for micro in range(3):
            opt.set_accumulation_boundary(micro == 2)
            x, y = _batch(4000 + 10 * s + micro)
            (crit(model(x), y) / 3).backward()
        opt.step()
    # reference: single process, grads averaged over micro-steps
    ref = _make_model()
    ropt = torch.optim.SGD(ref.parameters(), lr=0.1)
    for s in range(2):
        ropt.zero_grad()
        for micro in range(3):
            x, y = _batch(4000 + 10 * s + micro)
            (crit(ref(x), y) / 3).backward()
        ropt.step()
    flat = torch.cat([p.data.view(-1) for p in model.parameters()])
    rflat = torch.cat([p.data.view(-1) for p in ref.parameters()])
    assert torch.allclose(flat, rflat, atol=1e-5), \
        (flat - rflat).abs().max().item()


def test_dense_overlap_accumulation_world2():
    run_in_procs(_worker_overlap_with_accumulation, 2)


def test_adam_inner_optimizer_generic_path():
    """Reference API wraps ANY torch optimizer — Adam uses the
    densify-into-.grad + inner step() fallback."""
    torch.manual_seed(0)
    model = _make_model()
    inner = torch.optim.Adam(model.parameters(), lr=1e-3)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.1))
    assert not opt._fused
    crit = nn.CrossEntropyLoss()
    p0 = torch.cat([p.data.view(-1) for p in model.parameters()]).clone()
    for s in range(3):
        x, y = _batch(5000 + s)
        opt.zero_grad()
        crit(model(x), y).backward()
        opt.step()
    p1 = torch.cat([p.data.view(-1) for p in model.parameters()])
    assert not torch.allclose(p0, p1)
    assert torch.isfinite(p1).all()


def test_multiple_param_groups_generic_path():
    torch.manual_seed(0)
    model = _make_model()
    inner = torch.optim.SGD([
        {"params": model.fc1.parameters(), "lr": 0.1},
        {"params": model.fc2.parameters(), "lr": 0.01},
    ], momentum=0.9)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.1))
    assert not opt._fused  # >1 group -> inner step
    crit = nn.CrossEntropyLoss()
    x, y = _batch(6000)
    opt.zero_grad()
    crit(model(x), y).backward()
    opt.step()
    assert opt.train_iter == 1


def test_err_callback_fires():
    model = _make_model()
    inner = torch.optim.SGD(model.parameters(), lr=0.1)
    seen = []
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.1),
                                  err_callback=seen.append)
    # corrupt state to force a failure inside synchronize
    opt.compression = None
    opt.zero_grad()
    p = inner.param_groups[0]["params"][0]
    (p ** 2).sum().backward()
    try:
        opt.step()
    except Exception:
        pass
    assert len(seen) == 1


def _worker_lstm_gloo(rank, world):
    from gtopkssgd_amd.trainer import DLTrainer
    tr = DLTrainer(rank=rank, size=world, dnn="lstm", dataset="ptb",
                   batch_size=4, device="cpu", amp=False, num_workers=0,
                   n_samples=32, model_kwargs={"vocab_size": 80,
                                               "hidden": 16})
    opt = gt.DistributedOptimizer(tr.optimizer, tr.net.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01))
    for _ in range(2):
        tr.train(1, dist_optimizer=opt)
        tr.update_model(dist_optimizer=opt)
    m = tr.test(0)
    assert "perplexity" in m and m["perplexity"] > 0


def test_lstm_gloo_world2():
    run_in_procs(_worker_lstm_gloo, 2, timeout=240)
