# -*- coding: utf-8 -*-
"""CPU-path tests for the ops layer and the native scaler semantics.

These exercise the reference implementations the HIP kernels are validated
against on the GPU tier (tests/test_gpu_kernels.py).
"""

import math

import pytest
import torch

from stoke import ops
from stoke.amp import StokeGradScaler
from stoke.ops.fused_adam import FusedAdamW


def test_fused_adamw_matches_torch_adamw():
    torch.manual_seed(0)
    shapes = [(37,), (8, 9), (4, 5, 6)]
    p_ref = [torch.randn(s, dtype=torch.float64) for s in shapes]
    p_a = [torch.nn.Parameter(p.clone().float()) for p in p_ref]
    p_b = [torch.nn.Parameter(p.clone().float()) for p in p_ref]
    opt_a = FusedAdamW(p_a, lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                       weight_decay=0.01)
    opt_b = torch.optim.AdamW(p_b, lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                              weight_decay=0.01)
    for step in range(5):
        torch.manual_seed(100 + step)
        grads = [torch.randn_like(p) for p in p_a]
        for pa, pb, g in zip(p_a, p_b, grads):
            pa.grad = g.clone()
            pb.grad = g.clone()
        opt_a.step()
        opt_b.step()
    for pa, pb in zip(p_a, p_b):
        assert torch.allclose(pa, pb, rtol=1e-5, atol=1e-7), \
            (pa - pb).abs().max()


def test_fused_adamw_bf16_master_path():
    torch.manual_seed(0)
    p32 = torch.nn.Parameter(torch.randn(64))
    p16 = torch.nn.Parameter(p32.detach().to(torch.bfloat16))
    o32 = torch.optim.AdamW([p32], lr=1e-2, weight_decay=0.0)
    o16 = FusedAdamW([p16], lr=1e-2, weight_decay=0.0)
    for step in range(3):
        g = torch.randn(64)
        p32.grad = g.clone()
        p16.grad = g.to(torch.bfloat16)
        o32.step()
        o16.step()
    # master copy tracks the fp32 trajectory within bf16-grad noise
    master = o16.state[p16]["master"]
    assert torch.allclose(master, p32.detach(), rtol=3e-2, atol=3e-3)
    assert torch.equal(p16.detach(), master.to(torch.bfloat16))


def test_multi_tensor_l2norm():
    ts = [torch.randn(10), torch.randn(3, 7)]
    got = ops.multi_tensor_l2norm(ts)
    want = torch.sqrt(sum(t.pow(2).sum() for t in ts))
    assert torch.allclose(got, want.reshape(1), rtol=1e-6)


def test_multi_tensor_clamp():
    ts = [torch.randn(100) * 10]
    ops.multi_tensor_clamp_(ts, 0.5)
    assert ts[0].abs().max() <= 0.5


def test_multi_tensor_unscale_detects_inf():
    g = torch.tensor([1.0, float("inf"), 3.0])
    found = torch.zeros(1)
    ops.multi_tensor_unscale_([g], torch.tensor([0.5]), found)
    assert found.item() == 1.0
    assert g[0].item() == 0.5


def test_scaler_growth_and_backoff():
    sc = StokeGradScaler(init_scale=4.0, growth_factor=2.0, backoff_factor=0.5,
                         growth_interval=2, device="cpu")
    p = torch.nn.Parameter(torch.ones(4))
    opt = torch.optim.SGD([p], lr=0.1)

    def step_once(make_inf=False):
        opt.zero_grad()
        loss = (p * 2).sum()
        sc.scale(loss).backward()
        if make_inf:
            p.grad[0] = float("inf")
        sc.step(opt)
        sc.update()

    assert sc.get_scale() == 4.0
    step_once()          # growth tracker 1
    assert sc.get_scale() == 4.0
    step_once()          # growth tracker hits interval -> scale *2
    assert sc.get_scale() == 8.0
    before = p.detach().clone()
    step_once(make_inf=True)  # overflow: skip step, backoff
    assert sc.get_scale() == 4.0
    assert torch.equal(p.detach(), before)  # step skipped


def test_scaler_unscales_grads():
    sc = StokeGradScaler(init_scale=8.0, device="cpu")
    p = torch.nn.Parameter(torch.ones(3))
    opt = torch.optim.SGD([p], lr=1.0)
    loss = (p * 3).sum()
    sc.scale(loss).backward()
    assert torch.allclose(p.grad, torch.full((3,), 24.0))
    sc.unscale_(opt)
    assert torch.allclose(p.grad, torch.full((3,), 3.0))
    with pytest.raises(RuntimeError):
        sc.unscale_(opt)  # double unscale forbidden


def test_scaler_state_dict_roundtrip():
    sc = StokeGradScaler(init_scale=32.0, device="cpu")
    sc._lazy_init()
    sd = sc.state_dict()
    sc2 = StokeGradScaler(init_scale=1.0, device="cpu")
    sc2.load_state_dict(sd)
    assert sc2.get_scale() == 32.0


def test_scaler_step_with_fused_optimizer_device_flag():
    sc = StokeGradScaler(init_scale=2.0, device="cpu")
    p = torch.nn.Parameter(torch.ones(8))
    opt = FusedAdamW([p], lr=0.1)
    loss = (p**2).sum()
    sc.scale(loss).backward()
    sc.step(opt)   # goes through found_inf kwarg path
    sc.update()
    assert not torch.equal(p.detach(), torch.ones(8))


# ------------------------------------------------------------- FusedSGD
def test_fused_sgd_matches_torch_sgd():
    from stoke.ops.fused_sgd import FusedSGD

    for momentum, dampening, nesterov, wd in [
        (0.0, 0.0, False, 0.0),
        (0.9, 0.0, False, 1e-4),
        (0.9, 0.1, False, 0.0),
        (0.9, 0.0, True, 1e-4),
    ]:
        torch.manual_seed(0)
        shapes = [(37,), (8, 9), (4, 5, 6)]
        init = [torch.randn(s) for s in shapes]
        p_a = [torch.nn.Parameter(t.clone()) for t in init]
        p_b = [torch.nn.Parameter(t.clone()) for t in init]
        opt_a = FusedSGD(p_a, lr=0.05, momentum=momentum,
                         dampening=dampening, nesterov=nesterov,
                         weight_decay=wd)
        opt_b = torch.optim.SGD(p_b, lr=0.05, momentum=momentum,
                                dampening=dampening, nesterov=nesterov,
                                weight_decay=wd)
        for step in range(5):
            torch.manual_seed(100 + step)
            grads = [torch.randn_like(t) for t in init]
            for pa, pb, g in zip(p_a, p_b, grads):
                pa.grad = g.clone()
                pb.grad = g.clone()
            opt_a.step()
            opt_b.step()
        for pa, pb in zip(p_a, p_b):
            assert torch.allclose(pa, pb, rtol=1e-5, atol=1e-7), \
                (momentum, dampening, nesterov, wd, (pa - pb).abs().max())


def test_fused_sgd_bf16_master_path():
    from stoke.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    p32 = torch.nn.Parameter(torch.randn(64))
    p16 = torch.nn.Parameter(p32.detach().to(torch.bfloat16))
    o32 = torch.optim.SGD([p32], lr=0.05, momentum=0.9)
    o16 = FusedSGD([p16], lr=0.05, momentum=0.9)
    for step in range(3):
        g = torch.randn(64)
        p32.grad = g.clone()
        p16.grad = g.to(torch.bfloat16)
        o32.step()
        o16.step()
    master = o16.state[p16]["master"]
    assert torch.allclose(master, p32.detach(), rtol=3e-2, atol=3e-3)
    assert torch.equal(p16.detach(), master.to(torch.bfloat16))


def test_fused_sgd_scaler_skip_on_inf():
    from stoke.ops.fused_sgd import FusedSGD

    p = torch.nn.Parameter(torch.ones(8))
    opt = FusedSGD([p], lr=0.1, momentum=0.9)
    p.grad = torch.ones(8)
    opt.step(found_inf=torch.ones(1))  # must skip
    assert torch.equal(p.detach(), torch.ones(8))
    opt.step(found_inf=torch.zeros(1))
    assert not torch.equal(p.detach(), torch.ones(8))


def test_fused_adamw_nvme_offload(tmp_path):
    """File-backed (NVMe-tier) state matches in-RAM offload/plain AdamW."""
    import os

    torch.manual_seed(0)
    init = torch.randn(257)
    p_a = torch.nn.Parameter(init.clone())
    p_b = torch.nn.Parameter(init.clone())
    o_a = FusedAdamW([p_a], lr=1e-2, weight_decay=0.01,
                     offload_state=True, offload_path=str(tmp_path))
    o_b = torch.optim.AdamW([p_b], lr=1e-2, weight_decay=0.01)
    for step in range(4):
        torch.manual_seed(step)
        g = torch.randn_like(init)
        p_a.grad = g.clone()
        p_b.grad = g.clone()
        o_a.step()
        o_b.step()
    assert torch.allclose(p_a, p_b, rtol=1e-5, atol=1e-7)
    # state really is file-backed
    files = [f for f in os.listdir(tmp_path) if f.startswith("adamw_state_")]
    assert len(files) == 2  # exp_avg + exp_avg_sq for the fp32 param
