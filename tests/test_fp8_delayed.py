# -*- coding: utf-8 -*-
"""fp8 fused-quantize kernels + delayed-scaling linear — hardware-validated.

The v2 vectorized quantizers (HW packed fp8 converts, dual-layout output)
are the default `convert_linears_to_fp8` path and took Llama-3-8B fp8 from
a 0.81x regression to a 1.28x win over bf16 (NOTES.md), so this file runs
in the plain `pytest -m gpu` tier."""

import pytest
import torch

pytestmark = [pytest.mark.gpu]


def _ext():
    from stoke import _C

    return _C


@pytest.mark.parametrize("kind,mx", [(0, 448.0), (1, 57344.0)])
def test_fp8_quant_matches_eager(kind, mx):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    x = (torch.randn(1000, 512, device="cuda") * 3).bfloat16().contiguous()
    scale = torch.tensor([0.01], device="cuda")
    amax = torch.zeros(1, device="cuda")
    y = _ext().fp8_quant(x, scale, amax, kind)
    torch.cuda.synchronize()
    dt = torch.float8_e4m3fn if kind == 0 else torch.float8_e5m2
    want = (x.float() / scale).clamp(-mx, mx).to(dt)
    assert torch.equal(y.view(torch.uint8), want.view(torch.uint8))
    assert abs(amax.item() - x.float().abs().max().item()) < 1e-2


def test_fp8_quant_t_layouts():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(1)
    M, N = 136, 96  # non-multiples of 32 exercise tile edges
    x = torch.randn(M, N, device="cuda").bfloat16().contiguous()
    scale = torch.tensor([0.02], device="cuda")
    amax = torch.zeros(1, device="cuda")
    y, yt = _ext().fp8_quant_t(x, scale, amax, 0)
    torch.cuda.synchronize()
    assert y.shape == (M, N) and yt.shape == (N, M)
    want = (x.float() / scale).clamp(-448, 448).to(torch.float8_e4m3fn)
    assert torch.equal(y.view(torch.uint8), want.view(torch.uint8))
    assert torch.equal(
        yt.view(torch.uint8), want.t().contiguous().view(torch.uint8)
    )
    assert abs(amax.item() - x.float().abs().max().item()) < 1e-2


def test_fp8_delayed_linear_numerics():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke.nn.fp8 import fp8_available
    from stoke.nn.fp8_delayed import FP8LinearDelayed

    if not fp8_available():
        pytest.skip("fp8 unavailable")
    torch.manual_seed(3)
    lin = FP8LinearDelayed(256, 512).cuda().bfloat16()
    x = torch.randn(64, 256, device="cuda").bfloat16().requires_grad_(True)
    # two steps: step 1 primes scales, step 2 runs fully delayed
    for _ in range(2):
        y = lin(x)
        y.float().pow(2).mean().backward()
    ref = torch.nn.functional.linear(x.float(), lin.weight.float(),
                                     lin.bias.float())
    rel = (y.float() - ref).abs().mean() / ref.abs().mean()
    assert rel < 0.1, f"fp8 delayed fwd rel err {rel}"
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert lin.weight.grad is not None


def test_fp8_training_stability():
    """Loss decreases and scales stay finite over 30 delayed-scaling steps."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke.nn.fp8 import fp8_available
    from stoke.nn import convert_linears_to_fp8

    if not fp8_available():
        pytest.skip("fp8 unavailable")
    torch.manual_seed(7)
    import torch.nn as nn

    m = nn.Sequential(nn.Linear(1024, 2048), nn.SiLU(),
                      nn.Linear(2048, 1024)).cuda().bfloat16()
    assert convert_linears_to_fp8(m) == 2
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    x = torch.randn(64, 1024, device="cuda").bfloat16()
    y = torch.randn(64, 1024, device="cuda").bfloat16()
    losses = []
    for _ in range(30):
        loss = (m(x) - y).float().pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)).tolist())
    assert losses[-1] < losses[0] * 0.7, losses[::6]
    for mod in m.modules():
        for name in ("_sx", "_sw", "_sdy"):
            if hasattr(mod, name):
                assert torch.isfinite(getattr(mod, name)).all()
