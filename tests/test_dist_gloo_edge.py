# -*- coding: utf-8 -*-
"""Distributed edge cases on gloo (world 2): SDDP gradient accumulation,
FSDP + activation checkpointing, OSS + pinned-host optimizer-state offload."""

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from tests.test_dist_gloo import _data, _model, free_port, init_gloo


def _sddp_accum_worker(rank, world, port, steps, accum):
    pg = init_gloo(rank, world, port)
    from contextlib import nullcontext

    from stoke.shard import OSSOptimizer, StokeSDDPModule

    model = _model()
    opt = OSSOptimizer(
        [p for p in model.parameters()], optim=torch.optim.Adam, pg=pg, lr=0.01
    )
    sddp = StokeSDDPModule(model, sharded_optimizer=opt, pg=pg)
    for step in range(steps):
        for micro in range(accum):
            x, y = _data(rank * 1000 + step * 17 + micro)
            cm = sddp.no_sync() if micro < accum - 1 else nullcontext()
            with cm:
                (nn.CrossEntropyLoss()(sddp(x), y) / accum).backward()
        sddp.finish_backward()
        opt.step()
        opt.zero_grad()
    # reference: single-process mean over every (rank, micro) microbatch
    ref_model = _model()
    ref_opt = torch.optim.Adam(ref_model.parameters(), lr=0.01)
    for step in range(steps):
        ref_opt.zero_grad()
        losses = [
            nn.CrossEntropyLoss()(
                ref_model(_data(r * 1000 + step * 17 + m)[0]),
                _data(r * 1000 + step * 17 + m)[1],
            )
            for r in range(world)
            for m in range(accum)
        ]
        (sum(losses) / (world * accum)).backward()
        ref_opt.step()
    for p, r in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p.detach(), r.detach(), atol=1e-5), \
            f"rank {rank}: SDDP accum mismatch {(p.detach()-r.detach()).abs().max()}"
    dist.destroy_process_group()


def test_sddp_grad_accum_no_sync():
    torch.multiprocessing.spawn(
        _sddp_accum_worker, args=(2, free_port(), 2, 3), nprocs=2, join=True
    )


def _fsdp_ckpt_act_worker(rank, world, port, steps):
    pg = init_gloo(rank, world, port)
    from stoke.nn import apply_activation_checkpointing
    from stoke.shard import StokeFSDPModule

    model = _model()
    n = apply_activation_checkpointing(model, min_params=100)
    assert n > 0
    fsdp = StokeFSDPModule(
        model, pg=pg, compute_dtype=torch.float32,
        reshard_after_forward=True, min_wrap_params=100,
    )
    opt = torch.optim.Adam(fsdp.parameters(), lr=0.01)
    for step in range(steps):
        x, y = _data(rank * 1000 + step)
        loss = nn.CrossEntropyLoss()(fsdp(x), y)
        loss.backward()
        fsdp.finish_backward()
        opt.step()
        opt.zero_grad()
    ref_model = _model()
    ref_opt = torch.optim.Adam(ref_model.parameters(), lr=0.01)
    for step in range(steps):
        ref_opt.zero_grad()
        losses = [
            nn.CrossEntropyLoss()(ref_model(_data(r * 1000 + step)[0]),
                                  _data(r * 1000 + step)[1])
            for r in range(world)
        ]
        (sum(losses) / world).backward()
        ref_opt.step()
    sd = fsdp.full_state_dict()
    for name, rp in ref_model.named_parameters():
        assert torch.allclose(sd[name], rp.detach(), atol=2e-5), \
            f"rank {rank}: FSDP+ckpt {name} off by {(sd[name]-rp.detach()).abs().max()}"
    dist.destroy_process_group()


def test_fsdp_with_activation_checkpointing():
    torch.multiprocessing.spawn(
        _fsdp_ckpt_act_worker, args=(2, free_port(), 3), nprocs=2, join=True
    )


def _oss_offload_worker(rank, world, port, steps):
    pg = init_gloo(rank, world, port)
    from stoke.ddp import StokeDDPModule
    from stoke.ops.fused_adam import FusedAdamW
    from stoke.shard import OSSOptimizer

    model = _model()
    ddp = StokeDDPModule(model, pg=pg)
    opt = OSSOptimizer(
        [p for p in model.parameters()], optim=FusedAdamW, pg=pg,
        lr=0.01, weight_decay=0.0, offload_state=True,
    )
    for step in range(steps):
        x, y = _data(rank * 1000 + step)
        nn.CrossEntropyLoss()(ddp(x), y).backward()
        ddp.finish_backward()
        opt.step()
        opt.zero_grad()
    # offloaded state stays on host for the inner optimizer's shard
    for p in opt.optim.param_groups[0]["params"]:
        st = opt.optim.state.get(p)
        if st:
            assert st["exp_avg"].device.type == "cpu"
    # parity vs non-offloaded OSS on the same data
    model2 = _model()
    ddp2 = StokeDDPModule(model2, pg=pg)
    opt2 = OSSOptimizer(
        [p for p in model2.parameters()], optim=FusedAdamW, pg=pg,
        lr=0.01, weight_decay=0.0,
    )
    for step in range(steps):
        x, y = _data(rank * 1000 + step)
        nn.CrossEntropyLoss()(ddp2(x), y).backward()
        ddp2.finish_backward()
        opt2.step()
        opt2.zero_grad()
    for p, q in zip(model.parameters(), model2.parameters()):
        assert torch.allclose(p.detach(), q.detach(), atol=1e-6), \
            f"rank {rank}: offload OSS diverged {(p.detach()-q.detach()).abs().max()}"
    dist.destroy_process_group()


def test_oss_with_offloaded_state():
    torch.multiprocessing.spawn(
        _oss_offload_worker, args=(2, free_port(), 3), nprocs=2, join=True
    )


def _unused_param_worker(rank, world, port):
    pg = init_gloo(rank, world, port)
    from stoke.ddp import StokeDDPModule

    class Branchy(nn.Module):
        def __init__(self):
            super().__init__()
            torch.manual_seed(0)
            self.used = nn.Linear(8, 4)
            self.never = nn.Linear(8, 4)  # produces no grad

        def forward(self, x):
            return self.used(x)

    model = Branchy()
    ddp = StokeDDPModule(model, pg=pg, bucket_cap_mb=1,
                         find_unused_parameters=True)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    for step in range(2):
        x, y = _data(rank * 7 + step)
        nn.CrossEntropyLoss()(ddp(x), y).backward()
        ddp.finish_backward()  # must not hang on `never`'s missing grads
        opt.step()
        opt.zero_grad()
    # used params stay synced; never-params unchanged and identical
    for p in model.parameters():
        flat = p.detach().reshape(-1).clone()
        ref = flat.clone()
        dist.broadcast(ref, src=0)
        assert torch.allclose(flat, ref, atol=1e-6)
    dist.destroy_process_group()


def test_ddp_unused_parameters_no_hang():
    torch.multiprocessing.spawn(
        _unused_param_worker, args=(2, free_port()), nprocs=2, join=True
    )


def test_ddp_accum_with_bucket_view():
    from tests.test_dist_gloo import _ddp_worker

    torch.multiprocessing.spawn(
        _ddp_worker, args=(2, free_port(), 2, 3, True), nprocs=2, join=True
    )
