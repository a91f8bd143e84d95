# -*- coding: utf-8 -*-
"""Activation checkpointing + CPU optimizer-state offload (CPU tier)."""

import pytest
import torch

from benchmarks.models import llama_tiny
from stoke.nn import apply_activation_checkpointing
from stoke.ops.fused_adam import FusedAdamW


def test_act_ckpt_matches_plain_grads():
    torch.manual_seed(0)
    m1 = llama_tiny()
    m2 = llama_tiny()
    m2.load_state_dict(m1.state_dict())
    n = apply_activation_checkpointing(m2, min_params=10_000)
    assert n > 0
    x = torch.randint(0, 1024, (2, 16))
    m1.train()
    m2.train()
    m1(x).float().pow(2).mean().backward()
    m2(x).float().pow(2).mean().backward()
    for (n1, p1), (_, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5), n1


def test_act_ckpt_eval_mode_skips_recompute():
    m = llama_tiny()
    apply_activation_checkpointing(m, min_params=10_000)
    m.eval()
    x = torch.randint(0, 1024, (1, 8))
    with torch.no_grad():
        out = m(x)
    assert out.shape == (1, 8, 1024)


def test_act_ckpt_idempotent():
    m = llama_tiny()
    n1 = apply_activation_checkpointing(m, min_params=10_000)
    n2 = apply_activation_checkpointing(m, min_params=10_000)
    assert n1 > 0
    # second pass finds the same modules already wrapped (marked), wraps none
    # at a NEW level (marked modules are re-counted but not double-wrapped)
    x = torch.randint(0, 1024, (1, 8))
    m.train()
    m(x).float().pow(2).mean().backward()  # no double-recompute explosion
    assert n2 == n1


def test_offload_state_parity_with_plain():
    torch.manual_seed(1)
    a = torch.nn.Sequential(torch.nn.Linear(8, 32), torch.nn.Linear(32, 8))
    b = torch.nn.Sequential(torch.nn.Linear(8, 32), torch.nn.Linear(32, 8))
    b.load_state_dict(a.state_dict())
    oa = FusedAdamW(a.parameters(), lr=1e-2, weight_decay=0.1)
    ob = FusedAdamW(b.parameters(), lr=1e-2, weight_decay=0.1,
                    offload_state=True)
    for _ in range(5):
        x = torch.randn(4, 8)
        a(x).pow(2).mean().backward()
        b(x).pow(2).mean().backward()
        oa.step()
        ob.step()
        oa.zero_grad()
        ob.zero_grad()
    for pa, pb in zip(a.parameters(), b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-7)
    # state stays on host in offload mode
    for p in b.parameters():
        st = ob.state[p]
        assert st["exp_avg"].device.type == "cpu"


def test_offload_requires_supporting_optimizer():
    from stoke import (
        DeepspeedConfig,
        DeepspeedOffloadOptimizerConfig,
        DeepspeedZeROConfig,
        Stoke,
        StokeOptimizer,
    )

    # wiring check only: offload + non-supporting optimizer raises in
    # build_optimizer (exercised through the facade on a CPU/deepspeed-less
    # path is not possible, so call the runner mapping directly)
    import inspect

    assert "offload_state" in inspect.signature(FusedAdamW).parameters
    assert "offload_state" not in inspect.signature(torch.optim.AdamW).parameters


@pytest.mark.gpu
def test_gpu_offload_state_parity():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(2)
    a = torch.nn.Linear(64, 64).cuda()
    b = torch.nn.Linear(64, 64).cuda()
    b.load_state_dict(a.state_dict())
    oa = FusedAdamW(a.parameters(), lr=1e-2)
    ob = FusedAdamW(b.parameters(), lr=1e-2, offload_state=True)
    for _ in range(3):
        x = torch.randn(8, 64, device="cuda")
        a(x).pow(2).mean().backward()
        b(x).pow(2).mean().backward()
        oa.step()
        ob.step()
        oa.zero_grad()
        ob.zero_grad()
    assert torch.allclose(a.weight, b.weight, atol=1e-6)
    for p in b.parameters():
        assert ob.state[p]["exp_avg"].device.type == "cpu"
        assert ob.state[p]["exp_avg"].is_pinned()
