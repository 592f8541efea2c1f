"""Fused NHWC BatchNorm kernels vs plain fp32 PyTorch (MI355X only)."""
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from gtopkssgd_amd.models import resnet50
from gtopkssgd_amd.ops.fusedbn import FusedBNAct

pytestmark = pytest.mark.gpu

DEV = "cuda:0"
CL = torch.channels_last


def _run_pair(c, shape, relu, with_z, dtype, atol):
    torch.manual_seed(0)
    fused = FusedBNAct(c, relu=relu).to(DEV)
    ref_bn = nn.BatchNorm2d(c).to(DEV)
    ref_bn.load_state_dict(fused.state_dict())

    xf = torch.randn(*shape, device=DEV)
    zf = torch.randn(*shape, device=DEV) if with_z else None
    x = xf.to(dtype).contiguous(memory_format=CL).requires_grad_(True)
    z = (zf.to(dtype).contiguous(memory_format=CL).requires_grad_(True)
         if with_z else None)
    x2 = xf.clone().requires_grad_(True)  # fp32 eager reference
    z2 = zf.clone().requires_grad_(True) if with_z else None

    before = FusedBNAct.fused_calls
    out = fused(x, residual=z)
    assert FusedBNAct.fused_calls == before + 1, "HIP path did not run"
    ref = ref_bn(x2)
    if with_z:
        ref = ref + z2
    if relu:
        ref = F.relu(ref)
    assert torch.allclose(out.float(), ref, atol=atol), \
        (out.float() - ref).abs().max().item()

    g = torch.randn(*shape, device=DEV)
    out.backward(g.to(dtype).contiguous(memory_format=CL))
    ref.backward(g)
    assert torch.allclose(x.grad.float(), x2.grad, atol=atol), \
        (x.grad.float() - x2.grad).abs().max().item()
    if with_z:
        assert torch.allclose(z.grad.float(), z2.grad, atol=atol)
    assert torch.allclose(fused.weight.grad, ref_bn.weight.grad,
                          atol=atol * 20), \
        (fused.weight.grad - ref_bn.weight.grad).abs().max().item()
    assert torch.allclose(fused.bias.grad, ref_bn.bias.grad, atol=atol * 20)
    assert torch.allclose(fused.running_mean, ref_bn.running_mean,
                          atol=atol)
    assert torch.allclose(fused.running_var, ref_bn.running_var, atol=atol)


@pytest.mark.parametrize("relu,with_z", [(False, False), (True, False),
                                         (True, True)])
def test_fused_bn_fp32_matches_eager(relu, with_z):
    _run_pair(64, (8, 64, 14, 14), relu, with_z, torch.float32, 2e-4)


@pytest.mark.parametrize("c,shape", [(16, (4, 16, 32, 32)),
                                     (256, (4, 256, 14, 14)),
                                     (2048, (2, 2048, 7, 7))])
def test_fused_bn_fp32_channel_sizes(c, shape):
    _run_pair(c, shape, True, True, torch.float32, 2e-4)


def test_fused_bn_bf16_close_to_fp32_ref():
    """bf16 path vs fp32 eager reference.  Elements whose pre-ReLU value
    sits at the rounding boundary (|preact| < 0.02) are excluded from
    the dx comparison: the bf16 output rounds the ReLU mask differently
    than fp32 — both are valid subgradients, not a defect."""
    torch.manual_seed(0)
    c, shape = 64, (8, 64, 14, 14)
    fused = FusedBNAct(c, relu=True).to(DEV)
    ref_bn = nn.BatchNorm2d(c).to(DEV)
    ref_bn.load_state_dict(fused.state_dict())
    xf = torch.randn(*shape, device=DEV)
    zf = torch.randn(*shape, device=DEV)
    x = xf.to(torch.bfloat16).contiguous(memory_format=CL).requires_grad_(True)
    z = zf.to(torch.bfloat16).contiguous(memory_format=CL).requires_grad_(True)
    x2 = xf.clone().requires_grad_(True)
    z2 = zf.clone().requires_grad_(True)
    out = fused(x, residual=z)
    pre = ref_bn(x2) + z2
    ref = F.relu(pre)
    safe = (pre.abs() > 2e-2).detach()
    assert torch.allclose(out.float()[safe], ref[safe], atol=5e-2)
    g = torch.randn(*shape, device=DEV)
    out.backward(g.to(torch.bfloat16).contiguous(memory_format=CL))
    ref.backward(g)
    assert torch.allclose(x.grad.float()[safe], x2.grad[safe], atol=5e-2), \
        (x.grad.float()[safe] - x2.grad[safe]).abs().max().item()
    assert torch.allclose(z.grad.float()[safe], z2.grad[safe], atol=5e-2)
    # dgamma/dbeta oracle honouring the FUSED kernel's own ReLU mask
    # (channel sums would otherwise integrate the boundary flips)
    mask = (out.detach().float() > 0)
    # the kernel consumes the bf16-rounded dy — mirror that
    dy_eff = g.to(torch.bfloat16).float() * mask
    mean = x.detach().float().mean(dim=(0, 2, 3), keepdim=True)
    var = x.detach().float().var(dim=(0, 2, 3), unbiased=False,
                                 keepdim=True)
    xhat = (x.detach().float() - mean) / torch.sqrt(var + fused.eps)
    dgamma_exp = (dy_eff * xhat).sum(dim=(0, 2, 3))
    dbeta_exp = dy_eff.sum(dim=(0, 2, 3))
    assert torch.allclose(fused.weight.grad, dgamma_exp, atol=5e-2), \
        (fused.weight.grad - dgamma_exp).abs().max().item()
    assert torch.allclose(fused.bias.grad, dbeta_exp, atol=5e-2), \
        (fused.bias.grad - dbeta_exp).abs().max().item()


def test_fused_bn_eval_mode():
    torch.manual_seed(1)
    fused = FusedBNAct(32, relu=True).to(DEV)
    ref = nn.BatchNorm2d(32).to(DEV)
    # give running stats some structure
    with torch.no_grad():
        fused.running_mean.uniform_(-1, 1)
        fused.running_var.uniform_(0.5, 2)
    ref.load_state_dict(fused.state_dict())
    fused.eval()
    ref.eval()
    x = torch.randn(4, 32, 8, 8, device=DEV)
    with torch.no_grad():
        out = fused(x.contiguous(memory_format=CL))
        expect = F.relu(ref(x))
    assert torch.allclose(out, expect, atol=1e-5), \
        (out - expect).abs().max().item()


def test_resnet50_with_fused_bn_trains():
    torch.manual_seed(0)
    model = resnet50().to(DEV).to(memory_format=CL)
    before = FusedBNAct.fused_calls
    x = torch.randn(4, 3, 224, 224, device=DEV).contiguous(
        memory_format=CL)
    with torch.autocast("cuda", torch.bfloat16):
        loss = model(x).square().mean()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    assert FusedBNAct.fused_calls - before == 53  # every BN fused
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()
