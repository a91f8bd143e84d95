# -*- coding: utf-8 -*-
"""GPU numerics tests: HIP kernels vs plain PyTorch fp32 references.

Every test here runs on a real MI355X (gfx950) and REQUIRES the native
extension — a silent eager fallback would defeat the point.
"""

import pytest
import torch

gpu = pytest.mark.gpu

pytestmark = gpu


@pytest.fixture(scope="module", autouse=True)
def require_gpu_and_ext():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from stoke import ops

    assert ops.has_ext(), (
        "native HIP extension stoke._C must be built on the GPU box "
        "(python setup.py build_ext --inplace)"
    )


def test_ext_arch():
    from stoke import _C

    assert _C._built_for == "gfx950"


def test_unscale_inf_check():
    from stoke import ops

    g1 = torch.randn(1 << 20, device="cuda")
    g2 = torch.randn(333, device="cuda")
    ref1, ref2 = g1.clone(), g2.clone()
    inv = torch.tensor([0.25], device="cuda")
    found = torch.zeros(1, device="cuda")
    ops.multi_tensor_unscale_([g1, g2], inv, found)
    torch.cuda.synchronize()
    assert found.item() == 0.0
    assert torch.allclose(g1, ref1 * 0.25)
    assert torch.allclose(g2, ref2 * 0.25)
    # inject a NaN mid-tensor
    g1[12345] = float("nan")
    ops.multi_tensor_unscale_([g1, g2], inv, found)
    torch.cuda.synchronize()
    assert found.item() == 1.0


def test_unscale_misaligned_view():
    from stoke import ops

    base = torch.randn((1 << 16) + 7, device="cuda")
    view = base[3:]  # 12-byte offset: exercises the scalar fallback path
    ref = view.clone()
    found = torch.zeros(1, device="cuda")
    ops.multi_tensor_unscale_([view], torch.tensor([2.0], device="cuda"), found)
    torch.cuda.synchronize()
    assert torch.allclose(view, ref * 2.0)
    assert found.item() == 0.0


@pytest.mark.parametrize("dtype", [torch.float16, torch.bfloat16])
def test_unscale_half_dtypes(dtype):
    """fp16/bf16 grads go through the fused kernel — no host-sync fallback."""
    from stoke import ops

    g1 = (torch.randn(1 << 18, device="cuda") * 4).to(dtype)
    g2 = (torch.randn(517, device="cuda") * 4).to(dtype)
    ref1, ref2 = g1.float(), g2.float()
    inv = torch.tensor([0.25], device="cuda")
    found = torch.zeros(1, device="cuda")
    ops.multi_tensor_unscale_([g1, g2], inv, found)
    torch.cuda.synchronize()
    assert found.item() == 0.0
    assert torch.allclose(g1.float(), (ref1 * 0.25).to(dtype).float())
    assert torch.allclose(g2.float(), (ref2 * 0.25).to(dtype).float())
    # misaligned view exercises the scalar tail
    base = (torch.randn((1 << 12) + 9, device="cuda")).to(dtype)
    view = base[3:]
    refv = view.float()
    ops.multi_tensor_unscale_([view], inv, found)
    torch.cuda.synchronize()
    assert torch.allclose(view.float(), (refv * 0.25).to(dtype).float())
    # inf detection
    g2[7] = float("inf")
    ops.multi_tensor_unscale_([g1, g2], inv, found)
    torch.cuda.synchronize()
    assert found.item() == 1.0


def test_l2norm_vs_torch():
    from stoke import ops

    ts = [torch.randn(n, device="cuda") for n in (17, 1 << 18, 4097)]
    got = ops.multi_tensor_l2norm(ts)
    want = torch.sqrt(sum(t.float().pow(2).sum() for t in ts))
    torch.cuda.synchronize()
    assert torch.allclose(got.reshape(()), want, rtol=1e-5)


def test_l2norm_bf16():
    from stoke import ops

    ts = [torch.randn(1 << 16, device="cuda").bfloat16()]
    got = ops.multi_tensor_l2norm(ts)
    want = torch.sqrt(ts[0].float().pow(2).sum())
    assert torch.allclose(got.reshape(()), want, rtol=1e-3)


def test_clamp():
    from stoke import ops

    t = torch.randn(1 << 20, device="cuda") * 5
    ops.multi_tensor_clamp_([t], 0.75)
    torch.cuda.synchronize()
    assert t.abs().max().item() <= 0.75


def test_scale():
    from stoke import ops

    t = torch.randn(12345, device="cuda")
    ref = t.clone()
    ops.multi_tensor_scale_([t], torch.tensor([0.125], device="cuda"))
    torch.cuda.synchronize()
    assert torch.allclose(t, ref * 0.125)


def test_fused_adamw_vs_torch_fp32():
    """HIP fused AdamW against torch.optim.AdamW on identical fp32 inputs."""
    from stoke.ops.fused_adam import FusedAdamW

    torch.manual_seed(0)
    shapes = [(1 << 16,), (513,), (33, 77), (3,)]
    init = [torch.randn(s, device="cuda") for s in shapes]
    p_a = [torch.nn.Parameter(t.clone()) for t in init]
    p_b = [torch.nn.Parameter(t.clone()) for t in init]
    opt_a = FusedAdamW(p_a, lr=3e-3, betas=(0.9, 0.95), eps=1e-8,
                       weight_decay=0.1)
    opt_b = torch.optim.AdamW(p_b, lr=3e-3, betas=(0.9, 0.95), eps=1e-8,
                              weight_decay=0.1)
    for step in range(10):
        torch.manual_seed(step)
        gs = [torch.randn_like(t) for t in init]
        for pa, pb, g in zip(p_a, p_b, gs):
            pa.grad = g.clone()
            pb.grad = g.clone()
        opt_a.step()
        opt_b.step()
    torch.cuda.synchronize()
    for pa, pb in zip(p_a, p_b):
        err = (pa - pb).abs().max().item()
        assert err < 1e-5, f"fused adamw deviates: {err}"


@pytest.mark.parametrize("momentum,dampening,nesterov,wd", [
    (0.0, 0.0, False, 0.0),
    (0.9, 0.0, False, 1e-4),
    (0.9, 0.1, False, 0.0),
    (0.9, 0.0, True, 1e-4),
])
def test_fused_sgd_vs_torch_fp32(momentum, dampening, nesterov, wd):
    """HIP fused SGD against torch.optim.SGD on identical fp32 inputs."""
    from stoke.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    shapes = [(1 << 16,), (513,), (33, 77), (3,)]
    init = [torch.randn(s, device="cuda") for s in shapes]
    p_a = [torch.nn.Parameter(t.clone()) for t in init]
    p_b = [torch.nn.Parameter(t.clone()) for t in init]
    opt_a = FusedSGD(p_a, lr=0.05, momentum=momentum, dampening=dampening,
                     nesterov=nesterov, weight_decay=wd)
    opt_b = torch.optim.SGD(p_b, lr=0.05, momentum=momentum,
                            dampening=dampening, nesterov=nesterov,
                            weight_decay=wd)
    for step in range(8):
        torch.manual_seed(step)
        gs = [torch.randn_like(t) for t in init]
        for pa, pb, g in zip(p_a, p_b, gs):
            pa.grad = g.clone()
            pb.grad = g.clone()
        opt_a.step()
        opt_b.step()
    torch.cuda.synchronize()
    for pa, pb in zip(p_a, p_b):
        err = (pa - pb).abs().max().item()
        assert err < 1e-5, f"fused sgd deviates: {err}"


def test_fused_sgd_bf16_master():
    """bf16 param + fp32 master SGD path vs fp32 torch.optim.SGD oracle."""
    from stoke.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    init = torch.randn(1 << 14, device="cuda")
    p32 = torch.nn.Parameter(init.clone())
    p16 = torch.nn.Parameter(init.clone().bfloat16())
    o32 = torch.optim.SGD([p32], lr=0.05, momentum=0.9)
    o16 = FusedSGD([p16], lr=0.05, momentum=0.9)
    for step in range(5):
        torch.manual_seed(step)
        g = torch.randn_like(init)
        p32.grad = g.clone()
        p16.grad = g.bfloat16()
        o32.step()
        o16.step()
    torch.cuda.synchronize()
    master = o16.state[p16]["master"]
    assert torch.allclose(master, p32.detach(), rtol=3e-2, atol=3e-3)
    assert torch.equal(p16.detach(), master.to(torch.bfloat16))


def test_fused_adamw_bf16_master():
    """bf16 param + fp32 master path vs an fp32 torch.optim.AdamW oracle."""
    from stoke.ops.fused_adam import FusedAdamW

    torch.manual_seed(0)
    init = torch.randn(1 << 14, device="cuda")
    p32 = torch.nn.Parameter(init.clone())
    p16 = torch.nn.Parameter(init.clone().bfloat16())
    o32 = torch.optim.AdamW([p32], lr=1e-2, weight_decay=0.01)
    o16 = FusedAdamW([p16], lr=1e-2, weight_decay=0.01)
    for step in range(5):
        torch.manual_seed(step)
        g = torch.randn_like(init)
        p32.grad = g.clone()
        p16.grad = g.bfloat16()
        o32.step()
        o16.step()
    torch.cuda.synchronize()
    master = o16.state[p16]["master"]
    assert torch.allclose(master, p32.detach(), rtol=3e-2, atol=3e-3)
    assert torch.equal(p16.detach(), master.bfloat16())


def test_fused_adamw_skips_on_found_inf():
    from stoke.ops.fused_adam import FusedAdamW

    p = torch.nn.Parameter(torch.ones(1024, device="cuda"))
    opt = FusedAdamW([p], lr=0.1)
    p.grad = torch.ones_like(p)
    found = torch.ones(1, device="cuda")  # flag set -> skip
    opt.step(found_inf=found)
    torch.cuda.synchronize()
    assert torch.equal(p.detach(), torch.ones(1024, device="cuda"))
    found.zero_()
    opt.step(found_inf=found)
    torch.cuda.synchronize()
    assert not torch.equal(p.detach(), torch.ones(1024, device="cuda"))


def test_amp_update_scale():
    from stoke import ops

    scale = torch.tensor([1024.0], device="cuda")
    tracker = torch.zeros(1, dtype=torch.int32, device="cuda")
    no_inf = torch.zeros(1, device="cuda")
    inf = torch.ones(1, device="cuda")
    ops.amp_update_scale_(scale, tracker, no_inf, 2.0, 0.5, 2)
    torch.cuda.synchronize()
    assert scale.item() == 1024.0 and tracker.item() == 1
    ops.amp_update_scale_(scale, tracker, no_inf, 2.0, 0.5, 2)
    torch.cuda.synchronize()
    assert scale.item() == 2048.0 and tracker.item() == 0
    ops.amp_update_scale_(scale, tracker, inf, 2.0, 0.5, 2)
    torch.cuda.synchronize()
    assert scale.item() == 1024.0 and tracker.item() == 0


def test_scaler_gpu_loop():
    """Full fp16 scaler loop on GPU: scale, backward, unscale, step, update."""
    from stoke.amp import StokeGradScaler
    from stoke.ops.fused_adam import FusedAdamW

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(), torch.nn.Linear(128, 8)
    ).cuda()
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    sc = StokeGradScaler(init_scale=2.0**14, device="cuda")
    x = torch.randn(32, 64, device="cuda")
    y = torch.randint(0, 8, (32,), device="cuda")
    losses = []
    for i in range(20):
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", torch.float16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        sc.scale(loss).backward()
        sc.step(opt)
        sc.update()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


def test_multi_tensor_many_tensors():
    """>MAX_TENSORS tensors and >MAX_BLOCKS chunks in one call."""
    from stoke import ops

    ts = [torch.randn(3000, device="cuda") for _ in range(60)]
    ts.append(torch.randn(20_000_000, device="cuda"))  # ~305 chunks
    refs = [t.clone() for t in ts]
    found = torch.zeros(1, device="cuda")
    ops.multi_tensor_unscale_(ts, torch.tensor([0.5], device="cuda"), found)
    torch.cuda.synchronize()
    for t, r in zip(ts, refs):
        assert torch.allclose(t, r * 0.5)
    assert found.item() == 0.0


def test_fused_cross_entropy_vs_torch():
    """Fused bf16 online-lse CE vs fp32 torch, values and gradients."""
    from stoke.nn import fused_cross_entropy

    torch.manual_seed(11)
    for N, V in [(64, 50257), (33, 1031), (8, 8)]:
        logits = (torch.randn(N, V, device="cuda") * 3).bfloat16()
        target = torch.randint(0, V, (N,), device="cuda")
        target[::5] = -100  # ignored positions
        a = logits.clone().requires_grad_(True)
        b = logits.float().detach().requires_grad_(True)
        la = fused_cross_entropy(a, target)
        lb = torch.nn.functional.cross_entropy(b, target, ignore_index=-100)
        assert abs(la.item() - lb.item()) / (abs(lb.item()) + 1e-6) < 2e-2, \
            (N, V, la.item(), lb.item())
        (la * 3.0).backward()
        (lb * 3.0).backward()
        err = (a.grad.float() - b.grad).abs().max().item()
        scale = b.grad.abs().max().item() + 1e-9
        assert err / scale < 0.08, (N, V, err, scale)
    # all-ignored edge: loss 0, grads 0
    t2 = torch.full((16,), -100, device="cuda", dtype=torch.long)
    l2 = torch.randn(16, 128, device="cuda").bfloat16().requires_grad_(True)
    out = fused_cross_entropy(l2, t2)
    out.backward()
    assert out.item() == 0.0 and torch.all(l2.grad == 0)


def test_per_loss_scaler_gpu():
    """Per-loss scalers on CUDA: HIP stash-unscale path, independent backoff."""
    from stoke.amp import StokePerLossScaler

    p = torch.nn.Parameter(torch.ones(64, device="cuda"))
    opt = torch.optim.SGD([p], lr=0.1)
    sc = StokePerLossScaler(init_scale=2.0**8, device="cuda")
    x = torch.ones(64, device="cuda")
    a = (p * x).sum()
    b = (p * x).sum() * float("inf")
    sc.backward_per_loss([a, b], opt, [p])
    before = p.detach().clone()
    sc.step(opt)
    sc.update()
    torch.cuda.synchronize()
    assert torch.equal(p.detach(), before)
    assert sc._loss_scales[0].item() == 2.0**8
    assert sc._loss_scales[1].item() == 2.0**7
    opt.zero_grad()
    sc.backward_per_loss([(p * x).sum(), (p * x).sum() * 2.0], opt, [p])
    assert torch.allclose(p.grad, torch.full((64,), 3.0, device="cuda"),
                          atol=1e-5)
    sc.step(opt)
    sc.update()
    torch.cuda.synchronize()
    assert not torch.equal(p.detach(), before)
