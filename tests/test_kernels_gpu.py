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


def test_ef_range_plus_finish_equals_ef_compress():
    """Per-bucket accumulate (arbitrary unaligned cuts) + finish must
    equal the one-shot fused ef_compress bitwise (same kernels, same
    element-wise adds — only the launch partitioning differs)."""
    torch.manual_seed(44)
    n, k = 1_000_003, 1000
    g = (torch.randn(n) * torch.exp(torch.randn(n))).to(DEV)
    res0 = torch.rand(n, device=DEV) * 0.1
    res_a = res0.clone()
    v_a, i_a = ops.ef_compress(res_a, g, k)
    res_b = res0.clone()
    cuts = [0, 7, 1234, 65_537, 400_001, 999_999, n]  # odd offsets
    ops.ef_hist_zero(res_b.device)
    for s, e in zip(cuts[:-1], cuts[1:]):
        ops.ef_accum_hist_range(res_b, g, s, e)
    v_b, i_b = ops.ef_compress_finish(res_b, k)
    torch.cuda.synchronize()
    assert torch.equal(res_a, res_b)
    assert set(i_a.tolist()) == set(i_b.tolist())
    # same values at the same indices (output order may differ)
    da = torch.zeros(n, device=DEV)
    db = torch.zeros(n, device=DEV)
    ops.densify_(da, i_a, v_a, 1.0)
    ops.densify_(db, i_b, v_b, 1.0)
    assert torch.equal(da, db)


def test_merge_round_matches_cpu_composition():
    """Fused merge_round_ (7 launches) vs the primitive composition on
    CPU (scatter+scatter+gather_zero+topk_among), same wire buffers."""
    torch.manual_seed(45)
    n, k = 100_000, 500
    for trial in range(3):
        a = torch.randn(n)
        b = torch.randn(n)
        ia = torch.topk(a.abs(), k).indices.to(torch.int32)
        ib = torch.topk(b.abs(), k).indices.to(torch.int32)

        def pack(idx, vals):
            buf = torch.empty(2 * k, dtype=torch.int32)
            buf[:k] = idx
            buf[k:].view(torch.float32).copy_(vals)
            return buf

        dense_cpu = torch.zeros(n)
        send_ref = pack(ia, a[ia.long()])
        mval, midx = ops.merge_round_(dense_cpu, send_ref,
                                      pack(ib, b[ib.long()]), k)
        send_gpu = pack(ia, a[ia.long()]).to(DEV)
        recv_gpu = pack(ib, b[ib.long()]).to(DEV)
        dense_gpu = torch.zeros(n, device=DEV)
        gval, gidx = ops.merge_round_(dense_gpu, send_gpu, recv_gpu, k)
        torch.cuda.synchronize()
        assert torch.count_nonzero(dense_gpu) == 0, "scratch not cleaned"
        ref = torch.zeros(n)
        ops.densify_(ref, midx, mval, 1.0)
        got = torch.zeros(n, device=DEV)
        ops.densify_(got, gidx, gval, 1.0)
        assert torch.allclose(got.cpu(), ref, atol=1e-6), trial


def test_sparse_overlap_select_launched_from_backward():
    """With overlap on, the select must be fully launched by the last
    post-accumulate hook — before step() is called."""
    torch.manual_seed(46)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 4)).to(DEV)
    inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.01),
                                  warmup_iters=0)
    assert opt._sparse_overlap, "overlap should arm on GPU with ext"
    x = torch.randn(32, 64, device=DEV)
    y = torch.randint(0, 4, (32,), device=DEV)
    opt.zero_grad()
    torch.nn.CrossEntropyLoss()(model(x), y).backward()
    assert opt._overlap_select is not None, \
        "select not launched during backward"
    assert opt._n_issued == len(opt._buckets)
    opt.step()
    assert opt._overlap_select is None  # consumed and reset


def test_sparse_overlap_matches_synchronous_path():
    """3 training steps with overlap on vs off: identical parameters,
    momentum and EF residual (same element-wise ops, different launch
    partitioning)."""
    results = []
    for overlap in (True, False):
        torch.manual_seed(47)
        model = torch.nn.Sequential(
            torch.nn.Linear(64, 256), torch.nn.ReLU(),
            torch.nn.Linear(256, 8)).to(DEV)
        inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9,
                                weight_decay=1e-4)
        opt = gt.DistributedOptimizer(
            inner, model.named_parameters(),
            compression=gt.compressors["gtopk"](0.01), warmup_iters=0,
            overlap_sparse=overlap)
        assert opt._sparse_overlap == overlap
        crit = torch.nn.CrossEntropyLoss()
        gen = torch.Generator().manual_seed(48)
        for _ in range(3):
            x = torch.randn(32, 64, generator=gen).to(DEV)
            y = torch.randint(0, 8, (32,), generator=gen).to(DEV)
            opt.zero_grad()
            crit(model(x), y).backward()
            opt.step()
        torch.cuda.synchronize()
        results.append((opt.flat_param.clone().cpu(),
                        opt.flat_momentum.clone().cpu(),
                        opt.compression.residuals["flat"].clone().cpu()))
    for a, b in zip(results[0], results[1]):
        assert torch.equal(a, b)
