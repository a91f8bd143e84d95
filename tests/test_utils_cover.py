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
