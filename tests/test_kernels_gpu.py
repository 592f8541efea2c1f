"""HIP kernel numerics vs plain PyTorch fp32 references (MI355X only).

Every test here is @pytest.mark.gpu; the CPU equivalents of these
properties live in test_ops.py.
"""
import pytest
import torch

import gtopkssgd_amd as gt
from gtopkssgd_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _keys(t: torch.Tensor) -> torch.Tensor:
    """The select kernels' bucket key: |fp32| bits >> 18."""
    return (t.view(torch.int32) & 0x7FFFFFFF) >> 18


def test_ext_loads():
    assert ops.ext_available()


def test_ef_compress_bin_exact_selection():
    torch.manual_seed(0)
    n, k = 1_000_000, 1000
    g = (torch.randn(n) * torch.exp(torch.randn(n) * 2)).to(DEV)
    res = torch.zeros(n, device=DEV)
    v, i = ops.ef_compress(res, g, k)
    torch.cuda.synchronize()
    assert v.numel() == k and i.numel() == k
    idx = i.long()
    assert idx.unique().numel() == k, "duplicate indices selected"
    # values match the accumulated gradient at the selected positions
    assert torch.equal(v, g[idx])
    # residual: zero at selected, untouched elsewhere
    assert torch.count_nonzero(res[idx]) == 0
    mask = torch.ones(n, dtype=torch.bool, device=DEV)
    mask[idx] = False
    assert torch.equal(res[mask], g[mask])
    # bin-exactness: every selected key >= every unselected key
    sel_min = _keys(v.abs()).min()
    unsel_max = _keys(res[mask].abs()).max()
    assert sel_min >= unsel_max, (sel_min.item(), unsel_max.item())


def test_ef_compress_exact_when_bins_distinct():
    """With one element per bucket the bucketed select IS exact top-k."""
    n, k = 2048, 100
    keys = torch.randperm(6000)[:n] + 1024  # distinct bins, normal range
    bits = (keys << 18) | torch.randint(0, 1 << 18, (n,))
    sign = torch.randint(0, 2, (n,)) << 31
    vals = (bits | sign).to(torch.int32).view(torch.float32).to(DEV)
    res = torch.zeros(n, device=DEV)
    v, i = ops.ef_compress(res, vals, k)
    ref = torch.topk(vals.abs(), k).indices
    assert set(i.tolist()) == set(ref.tolist())


def test_ef_compress_accumulates_residual():
    n, k = 10_000, 50
    res = torch.full((n,), 0.5, device=DEV)
    g = torch.full((n,), 0.25, device=DEV)
    v, i = ops.ef_compress(res, g, k)
    assert torch.allclose(v, torch.full((k,), 0.75, device=DEV))


def test_ef_compress_density_one():
    n = 4096
    g = torch.randn(n, device=DEV)
    res = torch.zeros(n, device=DEV)
    v, i = ops.ef_compress(res, g, n)
    assert torch.count_nonzero(res) == 0
    recon = torch.zeros(n, device=DEV)
    recon[i.long()] = v
    assert torch.equal(recon, g)


def test_gather_zero_dedup_gpu():
    dense = torch.zeros(100, device=DEV)
    dense[3] = 5.0
    dense[7] = -2.0
    idx = torch.tensor([3, 7, 3, 3], dtype=torch.int32, device=DEV)
    vals = ops.gather_zero(dense, idx)
    assert vals[0] == 5.0 and vals[1] == -2.0
    # exactly one occurrence of idx 3 got the value
    assert vals[2] == 0.0 and vals[3] == 0.0
    assert torch.count_nonzero(dense) == 0


def test_scatter_add_gpu():
    dense = torch.zeros(100, device=DEV)
    idx = torch.tensor([1, 1, 2], dtype=torch.int32, device=DEV)
    val = torch.tensor([1.0, 2.0, 3.0], device=DEV)
    ops.scatter_add_(dense, idx, val)
    assert dense[1] == 3.0 and dense[2] == 3.0


def test_topk_among_gpu():
    torch.manual_seed(1)
    m, k = 5000, 200
    val = (torch.randn(m) * torch.exp(torch.randn(m) * 3)).to(DEV)
    idx = torch.arange(m, dtype=torch.int32, device=DEV)
    v, i = ops.topk_among(val, idx, k)
    assert v.numel() == k
    sel = torch.zeros(m, dtype=torch.bool, device=DEV)
    sel[i.long()] = True
    sel_min = _keys(v.abs()).min()
    unsel_max = _keys(val[~sel].abs()).max()
    assert sel_min >= unsel_max


def test_readd_rejected_gpu():
    res = torch.zeros(1000, device=DEV)
    lidx = torch.tensor([1, 3, 5], dtype=torch.int32, device=DEV)
    lval = torch.tensor([1.0, 2.0, 3.0], device=DEV)
    gidx = torch.tensor([3, 9], dtype=torch.int32, device=DEV)
    ops.readd_rejected_(res, lidx, lval, gidx)
    assert res[1] == 1.0 and res[5] == 3.0 and res[3] == 0.0
    # marks buffer must be clean for reuse
    ops.readd_rejected_(res, lidx, lval, gidx)
    assert res[1] == 2.0


def test_densify_gpu():
    dense = torch.zeros(50, device=DEV)
    idx = torch.tensor([4, 4, 9], dtype=torch.int32, device=DEV)
    val = torch.tensor([2.0, 0.0, 1.0], device=DEV)
    ops.densify_(dense, idx, val, 0.5)
    assert dense[4] == 1.0 and dense[9] == 0.5


def test_sgd_step_matches_torch_gpu():
    torch.manual_seed(3)
    for momentum, wd, nesterov in [(0.0, 0.0, False), (0.9, 0.0, False),
                                   (0.9, 5e-4, False), (0.9, 1e-4, True)]:
        n = 100_003  # odd: exercises the float4 tail
        p = torch.randn(n, device=DEV)
        ref_param = torch.nn.Parameter(p.clone())
        mine = p.clone()
        mom = torch.zeros(n, device=DEV)
        opt = torch.optim.SGD([ref_param], lr=0.1, momentum=momentum,
                              weight_decay=wd, nesterov=nesterov)
        for _ in range(3):
            g = torch.randn(n, device=DEV)
            ref_param.grad = g.clone()
            opt.step()
            ops.sgd_step_(mine, mom, g.clone(), lr=0.1, momentum=momentum,
                          weight_decay=wd, nesterov=nesterov)
        assert torch.allclose(mine, ref_param.data, atol=1e-5), \
            (momentum, wd, nesterov)


def test_optimizer_end_to_end_gpu():
    """Full gTop-k pipeline on GPU, world=1: converges on a toy task and
    runs entirely through the HIP kernels."""
    torch.manual_seed(7)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 256), torch.nn.ReLU(),
        torch.nn.Linear(256, 2)).to(DEV)
    inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01))
    crit = torch.nn.CrossEntropyLoss()
    g = torch.Generator().manual_seed(11)
    w_true = torch.randn(64, generator=g).to(DEV)
    losses = []
    for _ in range(60):
        x = torch.randn(128, 64, generator=g).to(DEV)
        y = (x @ w_true > 0).long()
        opt.zero_grad()
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert sum(losses[-10:]) / 10 < 0.7 * sum(losses[:10]) / 10, losses[::10]


def test_gpu_density1_matches_dense_sgd():
    """rho=1 gTop-k == plain torch SGD (the anchor invariant), on GPU."""
    torch.manual_seed(9)
    results = []
    for use_gt in (True, False):
        torch.manual_seed(9)
        model = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 4)).to(DEV)
        inner = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                                weight_decay=1e-4)
        opt = gt.DistributedOptimizer(
            inner, model.named_parameters(),
            compression=gt.compressors["gtopk"](1.0)) if use_gt else inner
        crit = torch.nn.CrossEntropyLoss()
        gen = torch.Generator().manual_seed(13)
        for _ in range(5):
            x = torch.randn(16, 32, generator=gen).to(DEV)
            y = torch.randint(0, 4, (16,), generator=gen).to(DEV)
            opt.zero_grad()
            crit(model(x), y).backward()
            opt.step()
        results.append(torch.cat([p.data.view(-1)
                                  for p in model.parameters()]).cpu())
    assert torch.allclose(results[0], results[1], atol=1e-5), \
        (results[0] - results[1]).abs().max().item()


def test_ef_compress_matches_torch_topk_random():
    """Two-level (26-bit) select: on continuous random data the selected
    index set must equal torch.topk's exactly (a 26-bit magnitude tie at
    the k-boundary has ~0 probability)."""
    torch.manual_seed(42)
    for n, k in [(1_000_000, 1000), (250_000, 25_000), (4096, 4096)]:
        g = (torch.randn(n) * torch.exp(torch.randn(n))).to(DEV)
        res = torch.zeros(n, device=DEV)
        v, i = ops.ef_compress(res, g, k)
        ref = torch.topk(g.abs(), k).indices
        assert set(i.tolist()) == set(ref.tolist()), (n, k)


def test_topk_among_matches_torch_topk_random():
    torch.manual_seed(43)
    m, k = 51_114, 25_557  # 2k -> k at ResNet-50 rho=0.001 scale
    val = torch.randn(m).to(DEV)
    idx = torch.arange(m, dtype=torch.int32, device=DEV)
    v, i = ops.topk_among(val, idx, k)
    ref = torch.topk(val.abs(), k).indices
    assert set(i.tolist()) == set(ref.tolist())
