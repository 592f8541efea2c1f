"""TopKCompressor + error-feedback properties (CPU reference path)."""
import torch

from gtopkssgd_amd.compression import (GTopKCompressor, NoneCompressor,
                                       TopKCompressor, compressors)


def test_registry():
    assert compressors["topk"] is TopKCompressor
    assert compressors["gtopk"] is GTopKCompressor
    assert compressors["none"] is NoneCompressor


def test_none_compressor_identity():
    c = NoneCompressor()
    t = torch.randn(100)
    v, i = c.compress(t)
    assert v is t and i is None


def test_topk_selects_largest_magnitude():
    torch.manual_seed(0)
    c = TopKCompressor(density=0.1)
    g = torch.randn(1000)
    v, i = c.compress(g.clone(), "t")
    assert v.numel() == 100 and i.dtype == torch.int32
    # selected set == torch.topk reference on |g| (fresh residual -> acc==g)
    ref = set(torch.topk(g.abs(), 100).indices.tolist())
    assert set(i.tolist()) == ref
    assert torch.allclose(v, g[i.long()])


def test_residual_conservation():
    """grad mass = selected values + residual after compress."""
    torch.manual_seed(1)
    c = TopKCompressor(density=0.05)
    total_grad = torch.zeros(500)
    for step in range(5):
        g = torch.randn(500)
        total_grad += g
        v, i = c.compress(g, "t")
        # selected + residual == sum of all grads so far minus previously
        # emitted values; track emitted mass:
        if step == 0:
            emitted = torch.zeros(500)
        emitted[i.long()] += v
        recon = emitted + c.residuals["t"]
        assert torch.allclose(recon, total_grad, atol=1e-5), step


def test_density_one_selects_everything():
    c = TopKCompressor(density=1.0)
    g = torch.randn(64)
    v, i = c.compress(g.clone(), "t")
    assert v.numel() == 64
    assert torch.allclose(v[i.long().argsort()],
                          g[i.long().sort().values.long()])
    assert torch.count_nonzero(c.residuals["t"]) == 0


def test_add_residuals_readds_rejected():
    torch.manual_seed(2)
    c = TopKCompressor(density=0.1)
    g = torch.randn(100)
    v, i = c.compress(g.clone(), "t")
    # pretend the global merge kept only the first 5 of our 10 picks
    kept = i[:5]
    rejected = i[5:]
    c.add_residuals(kept, "t")
    res = c.residuals["t"]
    assert torch.allclose(res[rejected.long()], v[5:])
    assert torch.all(res[kept.long()] == 0)


def test_state_dict_roundtrip():
    c = TopKCompressor(density=0.2)
    c.compress(torch.randn(50), "a")
    sd = c.state_dict()
    c2 = TopKCompressor(density=0.2)
    c2.load_state_dict(sd)
    assert torch.allclose(c2.residuals["a"], c.residuals["a"])
