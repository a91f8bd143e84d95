# -*- coding: utf-8 -*-
"""Fused RoPE: eager parity (CPU) + HIP numerics and gradient (GPU)."""

import pytest
import torch

from stoke.nn.rope import _eager_rope, apply_rope


def _cache(seq, hd, device="cpu", base=10000.0):
    inv = 1.0 / (base ** (torch.arange(0, hd, 2, device=device).float() / hd))
    t = torch.arange(seq, device=device).float()
    f = torch.outer(t, inv)
    return f.cos(), f.sin()


def test_cpu_rope_rotation_properties():
    torch.manual_seed(0)
    B, S, H, Dh = 2, 16, 4, 32
    x = torch.randn(B, S, H, Dh)
    cos, sin = _cache(S, Dh)
    y = apply_rope(x, cos, sin)
    # rotation preserves pair norms
    nx = (x[..., 0::2] ** 2 + x[..., 1::2] ** 2)
    ny = (y[..., 0::2] ** 2 + y[..., 1::2] ** 2)
    assert torch.allclose(nx, ny, atol=1e-5)
    # position 0 is the identity
    assert torch.allclose(y[:, 0], x[:, 0], atol=1e-6)
    # conj undoes the rotation
    back = _eager_rope(y, cos, sin, conj=True)
    assert torch.allclose(back, x, atol=1e-5)


@pytest.mark.gpu
@pytest.mark.parametrize("B,S,H,Dh", [(2, 128, 8, 128), (1, 64, 10, 64)])
def test_gpu_rope_vs_eager(B, S, H, Dh):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(1)
    x = torch.randn(B, S, H, Dh, device="cuda").bfloat16()
    cos, sin = _cache(S, Dh, device="cuda")
    got = apply_rope(x, cos, sin)
    want = _eager_rope(x.float(), cos, sin)
    err = (got.float() - want).abs().max().item()
    assert err < 0.03, f"rope err {err}"


@pytest.mark.gpu
def test_gpu_rope_backward_is_inverse_rotation():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(2)
    B, S, H, Dh = 2, 64, 4, 128
    x16 = torch.randn(B, S, H, Dh, device="cuda").bfloat16().requires_grad_(True)
    cos, sin = _cache(S, Dh, device="cuda")
    gy = torch.randn(B, S, H, Dh, device="cuda")
    apply_rope(x16, cos, sin).backward(gy.bfloat16())

    x32 = x16.detach().float().requires_grad_(True)
    _eager_rope(x32, cos, sin).backward(gy)
    err = (x16.grad.float() - x32.grad).abs().max().item()
    assert err / (x32.grad.abs().max().item() + 1e-6) < 0.03, err


# ------------------------------------------------------------------ SwiGLU
def test_cpu_swiglu_matches_eager():
    import torch.nn.functional as F

    from stoke.nn import swiglu

    torch.manual_seed(0)
    g = torch.randn(64, 32, requires_grad=True)
    u = torch.randn(64, 32, requires_grad=True)
    y = swiglu(g, u)
    assert torch.allclose(y, F.silu(g) * u, atol=1e-6)
    y.pow(2).sum().backward()
    assert g.grad is not None and u.grad is not None


@pytest.mark.gpu
def test_gpu_swiglu_vs_fp32():
    import torch.nn.functional as F

    from stoke.nn import swiglu

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(3)
    g16 = torch.randn(1000, 256, device="cuda").bfloat16().requires_grad_(True)
    u16 = torch.randn(1000, 256, device="cuda").bfloat16().requires_grad_(True)
    gy = torch.randn(1000, 256, device="cuda")
    swiglu(g16, u16).backward(gy.bfloat16())

    g32 = g16.detach().float().requires_grad_(True)
    u32 = u16.detach().float().requires_grad_(True)
    (F.silu(g32) * u32).backward(gy)
    for got, want in ((g16.grad, g32.grad), (u16.grad, u32.grad)):
        err = (got.float() - want).abs().max().item()
        assert err / (want.abs().max().item() + 1e-6) < 0.05, err


# --------------------------------------------------------------------- FP8
@pytest.mark.gpu
def test_gpu_fp8_linear_if_available():
    from stoke.nn import FP8Linear, fp8_available

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    if not fp8_available():
        pytest.skip("fp8 _scaled_mm unavailable on this build")
    torch.manual_seed(4)
    lin = FP8Linear(256, 512, bias=True).cuda().bfloat16()
    x = torch.randn(64, 256, device="cuda").bfloat16().requires_grad_(True)
    y = lin(x)
    assert y.shape == (64, 512) and y.dtype == torch.bfloat16
    # fp8 per-tensor scaling keeps relative error within a few percent
    ref = torch.nn.functional.linear(x.float(), lin.weight.float(),
                                     lin.bias.float())
    rel = (y.float() - ref).abs().mean() / ref.abs().mean()
    assert rel < 0.08, f"fp8 fwd rel err {rel}"
    y.float().pow(2).mean().backward()
    assert x.grad is not None and lin.weight.grad is not None
    assert torch.isfinite(x.grad).all()
