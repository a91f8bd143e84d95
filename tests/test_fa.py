# -*- coding: utf-8 -*-
"""In-house CDNA4 flash attention — hardware-validated (round 2).

The probe tests validate the MFMA A/B/C fragment lane maps against torch
matmuls with asymmetric operands (a symmetric operand hides row/col swaps);
the fwd/bwd tests compare against fp32 SDPA.  These kernels are the DEFAULT
attention path for head-dim 64 (stoke/nn/attention.py), so this file runs
in the plain `pytest -m gpu` tier.
"""

import pytest
import torch

pytestmark = [pytest.mark.gpu]


def _ext():
    from stoke import _C

    return _C


def test_mfma_probe_layout():
    """Validates the A/B/C fragment lane maps the FA kernel assumes."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    # asymmetric operands (guide: symmetric B hides row/col swaps)
    A = (torch.arange(16 * 32, device="cuda").float().reshape(16, 32)
         % 7 - 3).bfloat16() * 0.25
    B = (torch.arange(32 * 16, device="cuda").float().reshape(32, 16)
         % 5 - 2).bfloat16() * 0.5
    D = torch.zeros(16, 16, device="cuda", dtype=torch.float32)
    _ext().mfma_probe(A, B, D)
    torch.cuda.synchronize()
    want = A.float() @ B.float()
    err = (D - want).abs().max().item()
    assert err < 1e-2, (
        f"MFMA fragment-layout mismatch (max err {err}): fix the lane maps "
        "in csrc/fa_fwd.hip before debugging attention"
    )


def test_mfma_probe32_layout():
    """Validates the 32x32x16 A/B/C lane maps the v1 kernel assumes."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    A = (torch.arange(32 * 16, device="cuda").float().reshape(32, 16)
         % 7 - 3).bfloat16() * 0.25
    B = (torch.arange(16 * 32, device="cuda").float().reshape(16, 32)
         % 5 - 2).bfloat16() * 0.5
    D = torch.zeros(32, 32, device="cuda", dtype=torch.float32)
    _ext().mfma_probe32(A, B, D)
    torch.cuda.synchronize()
    want = A.float() @ B.float()
    err = (D - want).abs().max().item()
    assert err < 1e-2, (
        f"32x32x16 fragment-layout mismatch (max err {err}): fix the lane "
        "maps in csrc/fa_fwd.hip before debugging attention"
    )


@pytest.mark.parametrize("B,H,HKV,S,D,causal", [
    (1, 2, 2, 64, 64, False),
    (1, 2, 2, 128, 128, True),
    (2, 4, 2, 96, 128, True),   # GQA + ragged S
    (1, 1, 1, 300, 64, True),
    (1, 2, 1, 1024, 64, True),  # multi-256-block path (v1 grid)
    (1, 2, 2, 500, 128, True),  # ragged across block boundary
])
def test_fa_fwd_vs_sdpa(B, H, HKV, S, D, causal):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(1)
    q = torch.randn(B, H, S, D, device="cuda").bfloat16().contiguous()
    k = torch.randn(B, HKV, S, D, device="cuda").bfloat16().contiguous()
    v = torch.randn(B, HKV, S, D, device="cuda").bfloat16().contiguous()
    out, lse = _ext().fa_fwd(q, k, v, causal)
    torch.cuda.synchronize()
    want = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=causal,
        enable_gqa=(H != HKV),
    )
    err = (out.float() - want).abs().max().item()
    scale = want.abs().max().item() + 1e-6
    assert err / scale < 0.05, f"fa_fwd err {err} (scale {scale})"
    # logsumexp sanity: finite, and consistent with a direct computation
    s = (q.float() @ k.float().repeat_interleave(H // HKV, dim=1).transpose(-1, -2)
         ) / (D ** 0.5)
    if causal:
        mask = torch.triu(torch.ones(S, S, device="cuda", dtype=torch.bool), 1)
        s = s.masked_fill(mask, float("-inf"))
    want_lse = torch.logsumexp(s, dim=-1)
    lerr = (lse - want_lse).abs().max().item()
    assert lerr < 0.05, f"lse err {lerr}"


@pytest.mark.parametrize("B,H,HKV,S,D,causal", [
    (1, 2, 2, 64, 64, True),
    (1, 2, 1, 128, 128, True),
    (1, 1, 1, 96, 64, False),
])
def test_fa_bwd_vs_sdpa(B, H, HKV, S, D, causal):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke.nn.attention import flash_attention

    torch.manual_seed(2)
    q = torch.randn(B, H, S, D, device="cuda").bfloat16().requires_grad_(True)
    k = torch.randn(B, HKV, S, D, device="cuda").bfloat16().requires_grad_(True)
    v = torch.randn(B, HKV, S, D, device="cuda").bfloat16().requires_grad_(True)
    gy = torch.randn(B, H, S, D, device="cuda")
    flash_attention(q, k, v, causal=causal).backward(gy.bfloat16())

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    torch.nn.functional.scaled_dot_product_attention(
        q32, k32, v32, is_causal=causal, enable_gqa=(H != HKV)
    ).backward(gy)
    for got, want, nm in ((q.grad, q32.grad, "dq"), (k.grad, k32.grad, "dk"),
                          (v.grad, v32.grad, "dv")):
        err = (got.float() - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 0.06, f"{nm} err {err} (scale {scale})"


def test_fa_strided_views_match_contiguous():
    """The model path feeds transposed projection views (BSHD storage);
    the kernels take them with zero copies and must match contiguous."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke.nn.attention import flash_attention

    torch.manual_seed(5)
    B, H, S, D = 2, 4, 512, 64
    qkv = torch.randn(B, S, 3 * H * D, device="cuda").bfloat16()
    q, k, v = qkv.split(H * D, dim=-1)
    qv = q.view(B, S, H, D).transpose(1, 2)   # strided [B,H,S,D] view
    kv_ = k.view(B, S, H, D).transpose(1, 2)
    vv = v.view(B, S, H, D).transpose(1, 2)
    for t in (qv, kv_, vv):
        assert not t.is_contiguous()
    out_v = flash_attention(qv, kv_, vv, causal=True)
    out_c = flash_attention(qv.contiguous(), kv_.contiguous(),
                            vv.contiguous(), causal=True)
    assert torch.equal(out_v.float(), out_c.float())
    # backward through the views
    qg = qv.detach().clone().requires_grad_(True)
    kg = kv_.detach().clone().requires_grad_(True)
    vg = vv.detach().clone().requires_grad_(True)
    flash_attention(qg, kg, vg, causal=True).sum().backward()
    assert all(t.grad is not None and torch.isfinite(t.grad).all()
               for t in (qg, kg, vg))


def test_fa_attention_wrapper_routes():
    """attention() uses the native kernels for bf16 CUDA and SDPA elsewhere."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke.nn.attention import attention

    torch.manual_seed(6)
    q = torch.randn(1, 2, 128, 64, device="cuda").bfloat16()
    k = torch.randn(1, 2, 128, 64, device="cuda").bfloat16()
    v = torch.randn(1, 2, 128, 64, device="cuda").bfloat16()
    out = attention(q, k, v, causal=True)
    want = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=True)
    assert (out.float() - want).abs().max().item() < 0.05
