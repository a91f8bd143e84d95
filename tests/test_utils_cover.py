# -*- coding: utf-8 -*-
"""Coverage for utils helpers and scaler corner cases."""

import torch
import torch.distributed as dist

from stoke.utils import ParamNormalize, place_data_on_gpu, zero_optimizer_grads


def test_place_data_nested_structures():
    cpu = torch.device("cpu")
    data = {
        "a": torch.randn(2, 2),
        "b": [torch.randn(3), (torch.randn(1), "passthrough")],
        "c": 42,
    }
    out = place_data_on_gpu(data, fp16=None, device=cpu)
    assert out["a"].device == cpu
    assert out["b"][1][1] == "passthrough"
    assert out["c"] == 42


def test_place_data_deepspeed_half_cast():
    cpu = torch.device("cpu")
    x = torch.randn(4, 4)
    y = torch.randint(0, 5, (4,))
    ox = place_data_on_gpu(x, fp16="deepspeed", device=cpu)
    oy = place_data_on_gpu(y, fp16="deepspeed", device=cpu)
    assert ox.dtype == torch.half  # floating inputs cast (reference utils.py:64-69)
    assert oy.dtype == torch.long  # integral inputs preserved


def test_zero_grads_fused_vs_plain():
    m1 = torch.nn.Linear(4, 4)
    m1(torch.randn(2, 4)).sum().backward()
    opt = torch.optim.SGD(m1.parameters(), lr=0.1)
    zero_optimizer_grads(opt)
    assert all(p.grad is None for p in m1.parameters())  # set_to_none

    class FusedSGD(torch.optim.SGD):
        pass

    m2 = torch.nn.Linear(4, 4)
    m2(torch.randn(2, 4)).sum().backward()
    fopt = FusedSGD(m2.parameters(), lr=0.1)
    zero_optimizer_grads(fopt)
    # "Fused" in the class name => zeroed in place, kept allocated
    assert all(
        p.grad is not None and p.grad.abs().sum() == 0 for p in m2.parameters()
    )

    class FusedPrefersNone(torch.optim.SGD):
        zero_grad_prefers_none = True

    m3 = torch.nn.Linear(4, 4)
    m3(torch.randn(2, 4)).sum().backward()
    popt = FusedPrefersNone(m3.parameters(), lr=0.1)
    zero_optimizer_grads(popt)
    assert all(p.grad is None for p in m3.parameters())


def test_param_normalize_values():
    assert ParamNormalize.MILLION.value == 1e6
    assert ParamNormalize.BILLION.value == 1e9


def _sharded_scaler_worker(rank, world, port):
    import os

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from stoke.amp import StokeGradScaler

    scaler = StokeGradScaler(init_scale=8.0, device="cpu", sharded=True)
    m = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    loss = m(torch.randn(2, 4)).sum()
    scaler.scale(loss).backward()
    if rank == 1:  # poison ONE rank's grads; sharded sync must skip ALL ranks
        for p in m.parameters():
            p.grad[...] = float("inf")
    before = [p.detach().clone() for p in m.parameters()]
    scaler.step(opt)
    scaler.update()
    for p, b in zip(m.parameters(), before):
        assert torch.equal(p.detach(), b), "step should be skipped on every rank"
    assert scaler.get_scale() == 4.0  # backoff applied everywhere
    dist.destroy_process_group()


def test_sharded_scaler_syncs_found_inf():
    from tests.test_dist_gloo import free_port

    torch.multiprocessing.spawn(
        _sharded_scaler_worker, args=(2, free_port()), nprocs=2, join=True
    )


# --------------------------------------------------- per-loss scaler (CPU)
def test_per_loss_scaler_independent_backoff():
    import torch
    from stoke.amp import StokePerLossScaler

    p = torch.nn.Parameter(torch.ones(8))
    opt = torch.optim.SGD([p], lr=0.1)
    sc = StokePerLossScaler(init_scale=2.0**8, device="cpu")
    x = torch.ones(8)

    def losses():
        a = (p * x).sum() * 1.0
        b = (p * x).sum() * float("inf")  # loss 1 overflows every step
        return [a, b]

    sc.backward_per_loss(losses(), opt, [p])
    before = p.detach().clone()
    sc.step(opt)
    sc.update()
    # inf in loss-1's contribution -> step skipped
    assert torch.equal(p.detach(), before)
    # per-loss: scale 0 untouched (growth pending), scale 1 backed off
    assert sc._loss_scales[0].item() == 2.0**8
    assert sc._loss_scales[1].item() == 2.0**7
    # finite-only losses step normally and grads are true units
    opt.zero_grad()
    a = (p * x).sum()
    sc.backward_per_loss([a, a * 2.0], opt, [p])
    assert torch.allclose(p.grad, torch.full((8,), 3.0), atol=1e-5)
    sc.step(opt)
    sc.update()
    assert not torch.equal(p.detach(), before)


def test_per_loss_scaler_state_roundtrip():
    import torch
    from stoke.amp import StokePerLossScaler

    sc = StokePerLossScaler(init_scale=4.0, device="cpu")
    sc._loss_state(1)
    sc._loss_scales[1].fill_(16.0)
    sd = sc.state_dict()
    sc2 = StokePerLossScaler(device="cpu")
    sc2.load_state_dict(sd)
    assert sc2._loss_scales[1].item() == 16.0


def test_horovod_adasum_raises():
    import pytest
    import torch
    from stoke import Stoke, StokeOptimizer, HorovodConfig

    # status requires CUDA for distributed; shim probes like the gloo tests
    torch.cuda.is_available = lambda: True
    torch.distributed.is_nccl_available = lambda: True
    try:
        import os

        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT="29631",
                          RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
        torch.cuda.set_device = lambda *a, **k: None
        torch.nn.Module.cuda = lambda self, *a, **k: self
        from stoke import DDPConfig

        model = torch.nn.Linear(4, 2)
        with pytest.raises(NotImplementedError, match="Adasum"):
            Stoke(
                model=model,
                optimizer=StokeOptimizer(optimizer=torch.optim.SGD,
                                         optimizer_kwargs={"lr": 0.1}),
                loss=torch.nn.MSELoss(),
                batch_size_per_device=2,
                gpu=True,
                distributed="horovod",
                configs=[DDPConfig(local_rank=0, backend="gloo"),
                         HorovodConfig(op="Adasum")],
                verbose=False,
            )
    finally:
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()


def test_git_version():
    import stoke
    from stoke._version import get_versions

    v = get_versions()
    assert v["version"] and stoke.__version__ == v["version"]


def test_per_loss_scaler_accum_keeps_found_inf():
    """An inf folded into grads by micro-batch 1 must still skip the step
    even when micro-batch 2 is clean (gradient accumulation)."""
    import torch
    from stoke.amp import StokePerLossScaler

    p = torch.nn.Parameter(torch.ones(4))
    opt = torch.optim.SGD([p], lr=0.1)
    sc = StokePerLossScaler(init_scale=4.0, device="cpu")
    bad = (p * torch.ones(4)).sum() * float("inf")
    sc.backward_per_loss([bad], opt, [p])
    good = (p * torch.ones(4)).sum()
    sc.backward_per_loss([good], opt, [p])
    before = p.detach().clone()
    sc.step(opt)
    sc.update()
    assert torch.equal(p.detach(), before), "step ran over inf-tainted grads"


def test_lazy_loss_arithmetic():
    import torch
    from stoke.utils import LazyLoss

    a = LazyLoss(torch.tensor([6.0]))
    b = LazyLoss(torch.tensor([2.0]))
    assert float(a) == 6.0 and a.item() == 6.0
    assert float(a + b) == 8.0 and float(a + 1.0) == 7.0 and float(1.0 + a) == 7.0
    assert float(a - b) == 4.0 and float(10.0 - a) == 4.0
    assert float(a * b) == 12.0 and float(0.5 * a) == 3.0
    assert float(a / b) == 3.0 and float(12.0 / a) == 2.0
    assert float(-a) == -6.0 and float(abs(LazyLoss(torch.tensor([-3.0])))) == 3.0
    assert round(a / 4, 2) == 1.5
    assert (a > b) and (b < a) and (a >= 6.0) and (a <= 6.0) and a == 6.0
    assert f"{a:.3f}" == "6.000"
    # deferred all-reduce semantics: div applied lazily at first use
    c = LazyLoss(torch.tensor([9.0]), div=3.0)
    assert float(c) == 3.0


def test_fused_cross_entropy_cpu_fallback_matches_torch():
    import torch
    import torch.nn.functional as F
    from stoke.nn import fused_cross_entropy

    torch.manual_seed(0)
    logits = torch.randn(3, 5, 17)
    tgt = torch.randint(0, 17, (3, 5))
    tgt[0, 0] = -100
    a = fused_cross_entropy(logits, tgt, ignore_index=-100)
    b = F.cross_entropy(logits.reshape(-1, 17).float(), tgt.reshape(-1),
                        ignore_index=-100)
    assert torch.allclose(a, b)
