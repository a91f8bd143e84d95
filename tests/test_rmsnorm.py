# -*- coding: utf-8 -*-
"""StokeRMSNorm: CPU fallback + HIP numerics vs fp32 eager reference."""

import pytest
import torch

from stoke.nn import StokeRMSNorm


def _ref(x, w, eps=1e-5):
    xf = x.float()
    xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return xf * w.float()


def test_cpu_fallback_matches_eager():
    torch.manual_seed(0)
    m = StokeRMSNorm(64)
    m.weight.data.uniform_(0.5, 1.5)
    x = torch.randn(4, 7, 64)
    got = m(x)
    assert torch.allclose(got, _ref(x, m.weight), atol=1e-6)


def test_cpu_backward_matches_autograd():
    torch.manual_seed(1)
    m = StokeRMSNorm(32)
    x = torch.randn(8, 32, requires_grad=True)
    m(x).pow(2).sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


@pytest.mark.gpu
@pytest.mark.parametrize("T,D", [(64, 256), (1000, 4096), (17, 5120)])
def test_gpu_rmsnorm_forward_vs_fp32(T, D):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    x = torch.randn(T, D, device="cuda").bfloat16()
    m = StokeRMSNorm(D).cuda().bfloat16()
    m.weight.data.uniform_(0.5, 1.5)
    got = m(x)
    assert got.dtype == torch.bfloat16
    want = _ref(x, m.weight)
    err = (got.float() - want).abs().max().item()
    assert err < 0.05, f"rmsnorm fwd err {err}"


@pytest.mark.gpu
@pytest.mark.parametrize("T,D", [(128, 512), (333, 4096)])
def test_gpu_rmsnorm_backward_vs_fp32(T, D):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(2)
    x16 = torch.randn(T, D, device="cuda").bfloat16().requires_grad_(True)
    m = StokeRMSNorm(D).cuda().bfloat16()
    m.weight.data.uniform_(0.5, 1.5)
    gy = torch.randn(T, D, device="cuda")
    out = m(x16)
    out.backward(gy.bfloat16())

    x32 = x16.detach().float().requires_grad_(True)
    w32 = m.weight.detach().float().requires_grad_(True)
    _ref(x32, w32, m.eps).backward(gy)

    dxe = (x16.grad.float() - x32.grad).abs().max().item()
    scale = x32.grad.abs().max().item() + 1e-6
    assert dxe / scale < 0.05, f"dx err {dxe} (scale {scale})"
    dwe = (m.weight.grad.float() - w32.grad).abs().max().item()
    wscale = w32.grad.abs().max().item() + 1e-6
    assert dwe / wscale < 0.05, f"dw err {dwe} (scale {wscale})"


@pytest.mark.gpu
def test_gpu_rmsnorm_3d_and_odd_rows():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    x = torch.randn(3, 33, 1024, device="cuda").bfloat16()
    m = StokeRMSNorm(1024).cuda().bfloat16()
    y = m(x)
    assert y.shape == x.shape
    err = (y.float() - _ref(x, m.weight)).abs().max().item()
    assert err < 0.05
