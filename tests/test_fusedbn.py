"""FusedBNAct: CPU fallback parity (this file) — GPU kernel numerics
live in test_fusedbn_gpu.py."""
import torch
import torch.nn as nn
import torch.nn.functional as F

from gtopkssgd_amd.ops.fusedbn import FusedBNAct


def _reference(x, z, bn, relu):
    out = bn(x)
    if z is not None:
        out = out + z
    if relu:
        out = F.relu(out)
    return out


def test_cpu_fallback_matches_eager_bn_add_relu():
    torch.manual_seed(0)
    for relu, with_z in [(False, False), (True, False), (True, True)]:
        fused = FusedBNAct(8, relu=relu)
        ref_bn = nn.BatchNorm2d(8)
        ref_bn.load_state_dict(fused.state_dict())
        x = torch.randn(4, 8, 5, 5, requires_grad=True)
        x2 = x.detach().clone().requires_grad_(True)
        z = torch.randn(4, 8, 5, 5, requires_grad=True) if with_z else None
        z2 = z.detach().clone().requires_grad_(True) if with_z else None
        out = fused(x, residual=z)
        ref = _reference(x2, z2, ref_bn, relu)
        assert torch.allclose(out, ref, atol=1e-6)
        g = torch.randn_like(out)
        out.backward(g)
        ref.backward(g)
        assert torch.allclose(x.grad, x2.grad, atol=1e-6)
        if with_z:
            assert torch.allclose(z.grad, z2.grad, atol=1e-6)
        assert torch.allclose(fused.weight.grad, ref_bn.weight.grad,
                              atol=1e-6)
        assert torch.allclose(fused.running_mean, ref_bn.running_mean,
                              atol=1e-6)
        assert torch.allclose(fused.running_var, ref_bn.running_var,
                              atol=1e-6)


def test_state_dict_keys_match_batchnorm():
    fused = FusedBNAct(4, relu=True)
    plain = nn.BatchNorm2d(4)
    assert set(fused.state_dict()) == set(plain.state_dict())


def test_eval_mode_cpu():
    fused = FusedBNAct(8, relu=True)
    fused.eval()
    x = torch.randn(2, 8, 4, 4)
    out = fused(x)
    assert (out >= 0).all()


def test_no_running_stats_fallback():
    fused = FusedBNAct(8, relu=True, track_running_stats=False)
    x = torch.randn(2, 8, 4, 4, requires_grad=True)
    out = fused(x)
    out.sum().backward()
    assert (out >= 0).all() and x.grad is not None


def test_momentum_none_fallback():
    """momentum=None must follow nn.BatchNorm2d's cumulative-moving-
    average running stats (and bump num_batches_tracked), not freeze."""
    torch.manual_seed(1)
    fused = FusedBNAct(8, momentum=None)
    ref = nn.BatchNorm2d(8, momentum=None)
    ref.load_state_dict(fused.state_dict())
    for _ in range(3):
        x = torch.randn(2, 8, 4, 4)
        out = fused(x)
        out_ref = ref(x)
        assert torch.allclose(out, out_ref, atol=1e-6)
    assert torch.allclose(fused.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(fused.running_var, ref.running_var, atol=1e-6)
    assert int(fused.num_batches_tracked) == int(ref.num_batches_tracked) == 3


def test_non_pow2_channels_fall_back(monkeypatch):
    # 24 channels: 24/8=3 not a power of two -> fused path must decline
    fused = FusedBNAct(24, relu=True)
    x = torch.randn(2, 24, 4, 4)
    assert not fused._fusable(x.cuda() if torch.cuda.is_available() else x,
                              None) or not torch.cuda.is_available()
    out = fused(x)
    assert out.shape == x.shape
