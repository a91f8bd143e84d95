# -*- coding: utf-8 -*-
"""StokeLayerNorm: eager parity (CPU) + HIP numerics vs fp32 (GPU)."""

import pytest
import torch
import torch.nn.functional as F

from stoke.nn import StokeLayerNorm


def test_cpu_matches_torch_layernorm():
    torch.manual_seed(0)
    m = StokeLayerNorm(64)
    ref = torch.nn.LayerNorm(64)
    m.load_state_dict(ref.state_dict())
    x = torch.randn(4, 7, 64, requires_grad=True)
    y = m(x)
    assert torch.allclose(y, ref(x), atol=1e-6)
    y.pow(2).sum().backward()
    assert x.grad is not None


@pytest.mark.gpu
@pytest.mark.parametrize("T,D", [(64, 256), (1000, 1024), (33, 4096)])
def test_gpu_layernorm_forward_vs_fp32(T, D):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    x = torch.randn(T, D, device="cuda").bfloat16()
    m = StokeLayerNorm(D).cuda().bfloat16()
    m.weight.data.uniform_(0.5, 1.5)
    m.bias.data.uniform_(-0.5, 0.5)
    got = m(x)
    assert got.dtype == torch.bfloat16
    want = F.layer_norm(x.float(), (D,), m.weight.float(), m.bias.float(),
                        m.eps)
    err = (got.float() - want).abs().max().item()
    assert err < 0.05, f"ln fwd err {err}"


@pytest.mark.gpu
def test_gpu_layernorm_backward_vs_fp32():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(1)
    T, D = 333, 1024
    x16 = torch.randn(T, D, device="cuda").bfloat16().requires_grad_(True)
    m = StokeLayerNorm(D).cuda().bfloat16()
    m.weight.data.uniform_(0.5, 1.5)
    gy = torch.randn(T, D, device="cuda")
    m(x16).backward(gy.bfloat16())

    x32 = x16.detach().float().requires_grad_(True)
    w32 = m.weight.detach().float().requires_grad_(True)
    b32 = m.bias.detach().float().requires_grad_(True)
    F.layer_norm(x32, (D,), w32, b32, m.eps).backward(gy)
    for got, want, nm in ((x16.grad, x32.grad, "dx"),
                          (m.weight.grad, w32.grad, "dw"),
                          (m.bias.grad, b32.grad, "db")):
        err = (got.float() - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 0.05, f"{nm} err {err} (scale {scale})"
