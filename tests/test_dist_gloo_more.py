# -*- coding: utf-8 -*-
"""More distributed hardening on gloo: FSDP gradient accumulation, OSS with
multiple param groups, world-size-independent checkpoint reload."""

import os

import torch
import torch.distributed as dist
import torch.nn as nn

from tests.test_dist_gloo import _data, _model, free_port, init_gloo


def _fsdp_accum_worker(rank, world, port, steps, accum):
    pg = init_gloo(rank, world, port)
    from contextlib import nullcontext

    from stoke.shard import StokeFSDPModule

    model = _model()
    fsdp = StokeFSDPModule(model, pg=pg, compute_dtype=torch.float32,
                           reshard_after_forward=True, min_wrap_params=100)
    opt = torch.optim.Adam(fsdp.parameters(), lr=0.01)
    for step in range(steps):
        for micro in range(accum):
            x, y = _data(rank * 1000 + step * 13 + micro)
            cm = fsdp.no_sync() if micro < accum - 1 else nullcontext()
            with cm:
                (nn.CrossEntropyLoss()(fsdp(x), y) / accum).backward()
        fsdp.finish_backward()
        opt.step()
        opt.zero_grad()
    ref_model = _model()
    ref_opt = torch.optim.Adam(ref_model.parameters(), lr=0.01)
    for step in range(steps):
        ref_opt.zero_grad()
        losses = [
            nn.CrossEntropyLoss()(
                ref_model(_data(r * 1000 + step * 13 + m)[0]),
                _data(r * 1000 + step * 13 + m)[1],
            )
            for r in range(world)
            for m in range(accum)
        ]
        (sum(losses) / (world * accum)).backward()
        ref_opt.step()
    sd = fsdp.full_state_dict()
    for name, rp in ref_model.named_parameters():
        assert torch.allclose(sd[name], rp.detach(), atol=2e-5), \
            f"rank {rank}: FSDP accum {name} off {(sd[name]-rp.detach()).abs().max()}"
    dist.destroy_process_group()


def test_fsdp_grad_accum_no_sync():
    torch.multiprocessing.spawn(
        _fsdp_accum_worker, args=(2, free_port(), 2, 2), nprocs=2, join=True
    )


def _oss_groups_worker(rank, world, port):
    pg = init_gloo(rank, world, port)
    from stoke.ddp import StokeDDPModule
    from stoke.shard import OSSOptimizer

    model = _model()
    ddp = StokeDDPModule(model, pg=pg)
    params = list(model.parameters())
    groups = [
        {"params": params[: len(params) // 2], "lr": 0.05},
        {"params": params[len(params) // 2 :], "lr": 0.005},
    ]
    opt = OSSOptimizer(groups, optim=torch.optim.SGD, pg=pg, lr=0.01)
    for step in range(3):
        x, y = _data(rank * 1000 + step)
        nn.CrossEntropyLoss()(ddp(x), y).backward()
        ddp.finish_backward()
        opt.step()
        opt.zero_grad()
    ref_model = _model()
    rparams = list(ref_model.parameters())
    ref_opt = torch.optim.SGD(
        [
            {"params": rparams[: len(rparams) // 2], "lr": 0.05},
            {"params": rparams[len(rparams) // 2 :], "lr": 0.005},
        ],
        lr=0.01,
    )
    for step in range(3):
        ref_opt.zero_grad()
        losses = [
            nn.CrossEntropyLoss()(ref_model(_data(r * 1000 + step)[0]),
                                  _data(r * 1000 + step)[1])
            for r in range(world)
        ]
        (sum(losses) / world).backward()
        ref_opt.step()
    for p, r in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p.detach(), r.detach(), atol=1e-5), \
            f"rank {rank}: per-group lr mismatch {(p.detach()-r.detach()).abs().max()}"
    dist.destroy_process_group()


def test_oss_multiple_param_groups():
    torch.multiprocessing.spawn(
        _oss_groups_worker, args=(2, free_port()), nprocs=2, join=True
    )


def _save_w2_worker(rank, world, port, tmpdir):
    pg = init_gloo(rank, world, port)
    from stoke import io_ops
    from stoke.ddp import StokeDDPModule
    from stoke.shard import OSSOptimizer

    class _R:
        rank = pg.rank

        def barrier(self):
            pg.barrier()

    model = _model()
    ddp = StokeDDPModule(model, pg=pg)
    opt = OSSOptimizer(
        [p for p in model.parameters()], optim=torch.optim.Adam, pg=pg, lr=0.01
    )
    for step in range(2):
        x, y = _data(rank * 31 + step)
        nn.CrossEntropyLoss()(ddp(x), y).backward()
        ddp.finish_backward()
        opt.step()
        opt.zero_grad()
    io_ops.save_checkpoint(
        runner=_R(), shard="oss", model=model, optimizer=opt,
        path=str(tmpdir), backward_step=2, grad_accum_step=0,
        optimizer_step=2, name="w2", status={}, verbose=False,
    )
    if rank == 0:
        torch.save([p.detach().clone() for p in model.parameters()],
                   os.path.join(str(tmpdir), "expected.pt"))
    dist.destroy_process_group()


def test_checkpoint_world2_loads_world1(tmp_path):
    """OSS checkpoints are world-size independent: saved at world 2,
    loadable by a plain single-process optimizer (reference contract,
    SURVEY.md 5.4)."""
    torch.multiprocessing.spawn(
        _save_w2_worker, args=(2, free_port(), str(tmp_path)), nprocs=2,
        join=True,
    )
    ckpt = torch.load(
        os.path.join(str(tmp_path), "stoke-w2-backward-step-2.pt"),
        weights_only=False,
    )
    model = _model()
    model.load_state_dict(ckpt["model_state_dict"])
    expected = torch.load(os.path.join(str(tmp_path), "expected.pt"),
                          weights_only=False)
    for p, e in zip(model.parameters(), expected):
        assert torch.equal(p.detach(), e)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    opt.load_state_dict(ckpt["optimizer_state_dict"])  # full, re-shardable
    # optimizer state covers every parameter
    assert len(ckpt["optimizer_state_dict"]["state"]) == len(expected)
    # and training can continue
    x, y = _data(123)
    nn.CrossEntropyLoss()(model(x), y).backward()
    opt.step()


def test_benchmark_model_shapes_cpu():
    from benchmarks.models import GPT2, Llama

    g = GPT2(vocab=128, d=64, nlayer=2, nh=4, max_seq=32)
    out = g(torch.randint(0, 128, (2, 16)))
    assert out.shape == (2, 16, 128)
    out.float().pow(2).mean().backward()

    m = Llama(vocab=64, d=64, nlayer=2, nh=4, nkv=2, ffn=128, max_seq=32)
    out = m(torch.randint(0, 64, (2, 16)))
    assert out.shape == (2, 16, 64)
    out.float().pow(2).mean().backward()


def _fsdp_save_w2_worker(rank, world, port, tmpdir):
    pg = init_gloo(rank, world, port)
    from stoke import io_ops
    from stoke.shard import StokeFSDPModule

    class _R:
        rank = pg.rank

        def barrier(self):
            pg.barrier()

    model = _model()
    fsdp = StokeFSDPModule(model, pg=pg, compute_dtype=torch.float32,
                           min_wrap_params=100)
    opt = torch.optim.Adam(fsdp.parameters(), lr=0.01)
    for step in range(2):
        x, y = _data(rank * 41 + step)
        nn.CrossEntropyLoss()(fsdp(x), y).backward()
        fsdp.finish_backward()
        opt.step()
        opt.zero_grad()
    io_ops.save_checkpoint(
        runner=_R(), shard="fsdp", model=fsdp, optimizer=opt,
        path=str(tmpdir), backward_step=2, grad_accum_step=0,
        optimizer_step=2, name="fw2", status={}, verbose=False,
    )
    sd = fsdp.full_state_dict()  # collective: every rank participates
    if rank == 0:
        torch.save(sd, os.path.join(str(tmpdir), "expected_full.pt"))
    dist.destroy_process_group()


def _fsdp_load_w1_worker(rank, world, port, tmpdir):
    pg = init_gloo(rank, world, port)  # world 1
    from stoke import io_ops
    from stoke.shard import StokeFSDPModule

    class _R:
        rank = pg.rank
        device_id = "cpu"

        def barrier(self):
            pg.barrier()

    model = _model()
    fsdp = StokeFSDPModule(model, pg=pg, compute_dtype=torch.float32,
                           min_wrap_params=100)
    opt = torch.optim.Adam(fsdp.parameters(), lr=0.01)
    io_ops.load_checkpoint(
        runner=_R(), shard="fsdp", model=fsdp, optimizer=opt, gpu=False,
        path=str(tmpdir), tag="stoke-fw2-backward-step-2.pt",
    )
    got = fsdp.full_state_dict()
    want = torch.load(os.path.join(str(tmpdir), "expected_full.pt"),
                      weights_only=False)
    for k in want:
        assert torch.allclose(got[k], want[k], atol=1e-6), k
    # training continues from the re-sharded optimizer state
    x, y = _data(7)
    nn.CrossEntropyLoss()(fsdp(x), y).backward()
    fsdp.finish_backward()
    opt.step()
    dist.destroy_process_group()


def test_fsdp_checkpoint_world2_loads_world1(tmp_path):
    torch.multiprocessing.spawn(
        _fsdp_save_w2_worker, args=(2, free_port(), str(tmp_path)), nprocs=2,
        join=True,
    )
    torch.multiprocessing.spawn(
        _fsdp_load_w1_worker, args=(1, free_port(), str(tmp_path)), nprocs=1,
        join=True,
    )


# ------------------------------------------------- SDDP fp16 reduce safety
def _sddp_fp16_worker(rank, world, port):
    import torch.nn as nn
    from tests.conftest import init_gloo
    from stoke.shard import OSSOptimizer, StokeSDDPModule

    pg = init_gloo(rank, world, port)
    torch.manual_seed(0)
    model = nn.Linear(8, 8, bias=False)
    opt = OSSOptimizer([p for p in model.parameters()],
                       optim=torch.optim.SGD, pg=pg, lr=0.0)
    sddp = StokeSDDPModule(model, sharded_optimizer=opt, pg=pg,
                           reduce_fp16=True)
    # Large-magnitude grads: the raw fp16 SUM over ranks would overflow
    # (2 x 40000 > 65504); the pre-divide keeps partials in range.
    x = torch.full((4, 8), 100.0)
    out = sddp(x)
    (out.sum() * 100.0).backward()
    sddp.finish_backward()
    owned = [p.grad for p in model.parameters() if p.grad is not None]
    if rank == sddp._buckets[0].owner:
        assert owned and all(torch.isfinite(g).all() for g in owned), \
            "fp16 reduce overflowed despite pre-divide"
        # mean semantics preserved: both ranks saw identical data
        ref = torch.autograd.grad(
            (model(x).sum() * 100.0), model.parameters()
        )
        for g, r in zip(owned, ref):
            assert torch.allclose(g.float(), r.float(), rtol=2e-2), \
                (g - r).abs().max()
    torch.distributed.destroy_process_group()


def test_sddp_fp16_reduce_prediv_no_overflow():
    from tests.conftest import free_port

    torch.multiprocessing.spawn(
        _sddp_fp16_worker, args=(2, free_port()), nprocs=2, join=True
    )


# ------------------------------------------- per-loss scaler + DDP engine
def _per_loss_ddp_worker(rank, world, port):
    import torch.nn as nn
    from tests.conftest import init_gloo
    from stoke.amp import StokePerLossScaler
    from stoke.ddp import StokeDDPModule

    pg = init_gloo(rank, world, port)
    torch.manual_seed(0)
    model = nn.Linear(8, 4)
    ddp = StokeDDPModule(model, pg=pg, bucket_cap_mb=1)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    sc = StokePerLossScaler(init_scale=8.0, device="cpu")
    params = list(model.parameters())
    torch.manual_seed(100 + rank)
    x = torch.randn(4, 8)
    y = torch.randint(0, 4, (4,))
    with ddp.no_sync():
        losses = [nn.CrossEntropyLoss()(ddp(x), y),
                  0.5 * nn.CrossEntropyLoss()(ddp(x), y)]
        sc.backward_per_loss(losses, opt, params)
    ddp.sync_existing_grads()
    # grads equal across ranks (all-reduced true-unit grads)
    for p in params:
        flat = p.grad.reshape(-1).clone()
        torch.distributed.broadcast(flat, src=0)
        assert torch.allclose(flat, p.grad.reshape(-1), atol=1e-6)
    sc.step(opt)
    sc.update()
    # params stay equal across ranks
    for p in params:
        flat = p.detach().reshape(-1).clone()
        torch.distributed.broadcast(flat, src=0)
        assert torch.allclose(flat, p.detach().reshape(-1), atol=1e-6)
    torch.distributed.destroy_process_group()


def test_per_loss_scaler_with_ddp_sync():
    from tests.conftest import free_port

    torch.multiprocessing.spawn(
        _per_loss_ddp_worker, args=(2, free_port()), nprocs=2, join=True
    )
