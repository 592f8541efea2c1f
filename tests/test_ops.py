"""Unit tests for the ops layer (CPU reference implementations).

The same properties are re-checked against the HIP kernels on GPU in
test_kernels_gpu.py.
"""
import torch

from gtopkssgd_amd import ops


def test_ef_compress_matches_topk():
    torch.manual_seed(0)
    res = torch.zeros(1000)
    g = torch.randn(1000)
    v, i = ops.ef_compress(res, g, 50)
    ref_idx = set(torch.topk(g.abs(), 50).indices.tolist())
    assert set(i.tolist()) == ref_idx
    assert torch.all(res[i.long()] == 0)
    mask = torch.ones(1000, dtype=torch.bool)
    mask[i.long()] = False
    assert torch.allclose(res[mask], g[mask])


def test_gather_zero_dedups():
    dense = torch.zeros(10)
    dense[3] = 5.0
    dense[7] = -2.0
    idx = torch.tensor([3, 7, 3, 3], dtype=torch.int32)
    vals = ops.gather_zero(dense, idx)
    assert vals[0] == 5.0 and vals[1] == -2.0
    assert vals[2] == 0.0 and vals[3] == 0.0
    assert torch.count_nonzero(dense) == 0


def test_scatter_add_accumulates():
    dense = torch.zeros(10)
    idx = torch.tensor([1, 1, 2], dtype=torch.int32)
    val = torch.tensor([1.0, 2.0, 3.0])
    ops.scatter_add_(dense, idx, val)
    assert dense[1] == 3.0 and dense[2] == 3.0


def test_topk_among():
    val = torch.tensor([0.1, -5.0, 3.0, 0.01, -0.5])
    idx = torch.tensor([10, 20, 30, 40, 50], dtype=torch.int32)
    v, i = ops.topk_among(val, idx, 2)
    assert set(i.tolist()) == {20, 30}


def test_readd_rejected():
    res = torch.zeros(10)
    lidx = torch.tensor([1, 3, 5], dtype=torch.int32)
    lval = torch.tensor([1.0, 2.0, 3.0])
    gidx = torch.tensor([3, 9], dtype=torch.int32)
    ops.readd_rejected_(res, lidx, lval, gidx)
    assert res[1] == 1.0 and res[5] == 3.0 and res[3] == 0.0


def test_densify_skips_zero_dups():
    dense = torch.zeros(10)
    idx = torch.tensor([4, 4], dtype=torch.int32)
    val = torch.tensor([2.0, 0.0])
    ops.densify_(dense, idx, val, 0.5)
    assert dense[4] == 1.0


def test_sgd_step_matches_torch_sgd():
    torch.manual_seed(3)
    for momentum, wd, nesterov in [(0.0, 0.0, False), (0.9, 0.0, False),
                                   (0.9, 5e-4, False), (0.9, 1e-4, True)]:
        p_ref = torch.randn(200)
        p = p_ref.clone()
        mom = torch.zeros(200)
        ref_param = torch.nn.Parameter(p_ref.clone())
        opt = torch.optim.SGD([ref_param], lr=0.1, momentum=momentum,
                              weight_decay=wd, nesterov=nesterov)
        for step in range(3):
            g = torch.randn(200)
            ref_param.grad = g.clone()
            opt.step()
            ops.sgd_step_(p, mom, g.clone(), lr=0.1, momentum=momentum,
                          weight_decay=wd, nesterov=nesterov)
            assert torch.allclose(p, ref_param.data, atol=1e-6), \
                (momentum, wd, nesterov, step)


def test_sgd_step_grad_scale_and_zero():
    p = torch.ones(10)
    mom = torch.zeros(10)
    g = torch.full((10,), 2.0)
    ops.sgd_step_(p, mom, g, lr=0.5, momentum=0.0, weight_decay=0.0,
                  grad_scale=0.5, zero_grad=True)
    assert torch.allclose(p, torch.full((10,), 0.5))
    assert torch.count_nonzero(g) == 0
