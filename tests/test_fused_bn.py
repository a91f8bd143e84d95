# -*- coding: utf-8 -*-
"""FusedBNAct2d: CPU fallback parity (here) + HIP numerics vs fp32 torch
reference (gpu tier)."""

import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from stoke.nn import FusedBNAct2d


def _ref_forward(x, bn, residual=None, relu=True, training=True):
    # NOTE: running stats update in-place on the module (no clone) so tests
    # can compare fused.running_* against ref.running_* after the call.
    out = F.batch_norm(
        x, bn.running_mean, bn.running_var, bn.weight, bn.bias,
        training, bn.momentum, bn.eps,
    )
    if residual is not None:
        out = out + residual
    return F.relu(out) if relu else out


def test_cpu_fallback_matches_torch_bn():
    torch.manual_seed(0)
    fused = FusedBNAct2d(8, relu=True)
    ref_bn = nn.BatchNorm2d(8)
    x = torch.randn(4, 8, 5, 5)
    idt = torch.randn(4, 8, 5, 5)
    got = fused(x, residual=idt)
    want = F.relu(ref_bn(x) + idt)
    assert torch.allclose(got, want, atol=1e-6)
    assert torch.allclose(fused.running_mean, ref_bn.running_mean, atol=1e-6)
    assert torch.allclose(fused.running_var, ref_bn.running_var, atol=1e-6)


def test_state_dict_compatible_with_batchnorm():
    fused = FusedBNAct2d(16)
    bn = nn.BatchNorm2d(16)
    bn.weight.data.uniform_()
    bn.bias.data.uniform_()
    bn.running_mean.uniform_()
    bn.running_var.uniform_(0.5, 1.5)
    fused.load_state_dict(bn.state_dict())
    assert torch.equal(fused.weight, bn.weight)
    assert torch.equal(fused.running_var, bn.running_var)


def test_cpu_backward_matches():
    torch.manual_seed(0)
    x1 = torch.randn(2, 8, 4, 4, requires_grad=True)
    x2 = x1.detach().clone().requires_grad_(True)
    i1 = torch.randn(2, 8, 4, 4, requires_grad=True)
    i2 = i1.detach().clone().requires_grad_(True)
    fused = FusedBNAct2d(8, relu=True)
    bn = nn.BatchNorm2d(8)
    fused(x1, residual=i1).pow(2).sum().backward()
    F.relu(bn(x2) + i2).pow(2).sum().backward()
    assert torch.allclose(x1.grad, x2.grad, atol=1e-5)
    assert torch.allclose(i1.grad, i2.grad, atol=1e-5)
    assert torch.allclose(fused.weight.grad, bn.weight.grad, atol=1e-5)


# ------------------------------------------------------------------ GPU tier
@pytest.mark.gpu
@pytest.mark.parametrize("C,HW,relu,res", [
    (64, 56, True, False),
    (256, 56, True, True),
    (2048, 7, False, False),
    (512, 28, True, True),
])
def test_gpu_fused_bn_forward_vs_fp32(C, HW, relu, res):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    N = 8
    x32 = torch.randn(N, C, HW, HW, device="cuda")
    idt32 = torch.randn(N, C, HW, HW, device="cuda") if res else None
    x16 = x32.bfloat16().to(memory_format=torch.channels_last)
    idt16 = idt32.bfloat16().to(memory_format=torch.channels_last) if res else None
    fused = FusedBNAct2d(C, relu=relu).cuda()
    ref = nn.BatchNorm2d(C).cuda()
    got = fused(x16, residual=idt16)
    want = _ref_forward(x32, ref, idt32, relu)
    assert got.dtype == torch.bfloat16
    err = (got.float() - want).abs().max().item()
    assert err < 0.05, f"fused BN fwd err {err}"
    assert torch.allclose(fused.running_mean, ref.running_mean, atol=1e-2)
    assert torch.allclose(fused.running_var, ref.running_var, atol=1e-2)


@pytest.mark.gpu
@pytest.mark.parametrize("relu,res", [(True, True), (True, False),
                                      (False, False)])
def test_gpu_fused_bn_backward_vs_fp32(relu, res):
    """fp32 reference consumes the SAME bf16-rounded input the kernel sees:
    with an unrounded fp32 input the ReLU mask flips at the boundary (bn
    output sign differs inside bf16 rounding), which turns an O(eps) input
    difference into an O(1) gradient difference at those positions — an
    input-rounding artifact, not a kernel defect."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(1)
    N, C, HW = 4, 128, 14
    x16 = (torch.randn(N, C, HW, HW, device="cuda").bfloat16()
           .to(memory_format=torch.channels_last).requires_grad_(True))
    i16 = (torch.randn(N, C, HW, HW, device="cuda").bfloat16()
           .to(memory_format=torch.channels_last).requires_grad_(True)
           if res else None)
    x32 = x16.detach().float().requires_grad_(True)
    i32 = i16.detach().float().requires_grad_(True) if res else None
    fused = FusedBNAct2d(C, relu=relu).cuda()
    ref = nn.BatchNorm2d(C).cuda()
    gy = torch.randn(N, C, HW, HW, device="cuda")
    out16 = fused(x16, residual=i16)
    out16.backward(gy.bfloat16().to(memory_format=torch.channels_last))
    out32 = _ref_forward(x32, ref, i32, relu)
    out32.backward(gy)
    gerr = (x16.grad.float() - x32.grad).abs().max().item()
    scale = x32.grad.abs().max().item() + 1e-6
    assert gerr / scale < 0.1, f"dx err {gerr} (scale {scale})"
    if res:
        ierr = (i16.grad.float() - i32.grad).abs().max().item()
        assert ierr / (i32.grad.abs().max().item() + 1e-6) < 0.05, ierr
    werr = (fused.weight.grad - ref.weight.grad).abs().max().item()
    wscale = ref.weight.grad.abs().max().item() + 1e-6
    assert werr / wscale < 0.05, f"dgamma err {werr}"
    berr = (fused.bias.grad - ref.bias.grad).abs().max().item()
    assert berr / (ref.bias.grad.abs().max().item() + 1e-6) < 0.05, berr


@pytest.mark.gpu
def test_gpu_fused_bn_eval_mode():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    C = 64
    fused = FusedBNAct2d(C, relu=True).cuda().eval()
    fused.running_mean.uniform_(-0.5, 0.5)
    fused.running_var.uniform_(0.5, 1.5)
    x32 = torch.randn(2, C, 8, 8, device="cuda")
    x16 = x32.bfloat16().to(memory_format=torch.channels_last)
    with torch.no_grad():
        got = fused(x16)
        want = F.relu(F.batch_norm(
            x32, fused.running_mean, fused.running_var, fused.weight,
            fused.bias, False, 0.1, fused.eps,
        ))
    assert (got.float() - want).abs().max().item() < 0.05


@pytest.mark.gpu
def test_gpu_resnet_block_fused_vs_eager():
    """A whole Bottleneck with fused BN vs the eager composition, bf16."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from benchmarks.models import Bottleneck, _Downsample

    torch.manual_seed(0)
    blk = Bottleneck(64, 64, stride=1,
                     downsample=_Downsample(64, 256, 1)).cuda()
    blk = blk.to(memory_format=torch.channels_last)
    x = torch.randn(4, 64, 32, 32, device="cuda").to(
        memory_format=torch.channels_last
    )
    with torch.autocast("cuda", torch.bfloat16):
        y = blk(x)
        y.float().pow(2).mean().backward()
    assert y.dtype == torch.bfloat16
    assert all(
        p.grad is not None and torch.isfinite(p.grad).all()
        for p in blk.parameters()
    )
