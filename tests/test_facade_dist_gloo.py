# -*- coding: utf-8 -*-
"""Full-facade distributed test on gloo (world size 2, CPU).

The engine-level gloo tests (test_dist_gloo.py) cover DDP/OSS/SDDP/FSDP in
isolation; THIS test drives the whole ``Stoke`` facade -> runner ->
engine -> io path under ``distributed="ddp"`` so the exact code the driver
runs at N>1 on the GPU node is exercised here first.  GPU/NCCL availability
probes are patched inside each spawned worker (gloo backend, CPU tensors,
``.cuda()`` as identity) — the orchestration logic is identical either way.
"""

import os

import pytest
import torch
import torch.distributed as dist

from tests.test_dist_gloo import free_port


def _patch_gpu_probes():
    import stoke.status as status_mod

    torch.cuda.is_available = lambda: True  # status probe
    torch.distributed.is_nccl_available = lambda: True
    torch.cuda.set_device = lambda *_a, **_k: None
    torch.cuda.current_device = lambda: 0
    torch.cuda.is_current_stream_capturing = lambda: False  # optimizer check
    torch.nn.Module.cuda = lambda self, *a, **k: self  # stay on CPU
    _orig_load = torch.load
    torch.load = lambda *a, **k: _orig_load(
        *a, **{**k, "map_location": "cpu"}  # no real device in this worker
    )
    assert status_mod is not None


def _facade_worker(rank, world, port, tmpdir, mode):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        RANK=str(rank),
        WORLD_SIZE=str(world),
        LOCAL_RANK=str(rank),
    )
    _patch_gpu_probes()

    import torch.nn as nn

    from stoke import DDPConfig, FairscaleOSSConfig, Stoke, StokeOptimizer

    torch.manual_seed(10 + rank)  # startup broadcast must equalize
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    kw = dict(
        gpu=True,
        fp16=None,
        distributed="ddp",
        configs=[DDPConfig(local_rank=rank, backend="gloo", bucket_cap_mb=1),
                 FairscaleOSSConfig()],
        verbose=False,
    )
    if mode == "oss":
        kw["fairscale_oss"] = True
    s = Stoke(
        model=model,
        optimizer=StokeOptimizer(
            optimizer=torch.optim.AdamW, optimizer_kwargs={"lr": 1e-2}
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        grad_accum_steps=2,
        **kw,
    )
    assert s.world_size == world
    assert s.rank == rank
    # params equal across ranks after startup broadcast
    flat = torch.cat([p.detach().reshape(-1) for p in s.model_access.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref), "startup param sync failed"

    torch.manual_seed(100 + rank)  # different data per rank
    for step in range(4):
        for micro in range(2):
            x = torch.randn(4, 8)
            y = torch.randint(0, 4, (4,))
            out = s.model(x)
            loss = s.loss(out, y)
            s.backward(loss)
            s.step()
    assert s._optimizer_steps == 4, s._optimizer_steps
    assert s._backward_steps == 8
    # grads synced -> params identical on every rank
    flat = torch.cat([p.detach().reshape(-1) for p in s.model_access.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref, atol=1e-6), "post-train param divergence"

    # save/load round-trip through the facade (rank-0 write behind fences)
    path, tag = s.save(path=str(tmpdir), name="facade")
    extras = s.load(path=path, tag=tag)
    assert extras is None or isinstance(extras, dict)
    assert s._optimizer_steps == 4  # counters restored
    loss_val = s.detach_and_sync_loss(torch.tensor(float(rank)))
    assert abs(loss_val - 0.5) < 1e-6  # mean over ranks 0,1
    dist.destroy_process_group()


@pytest.mark.parametrize("mode", ["ddp", "oss"])
def test_facade_distributed_gloo(tmp_path, mode):
    torch.multiprocessing.spawn(
        _facade_worker, args=(2, free_port(), str(tmp_path), mode),
        nprocs=2, join=True,
    )


def _hvd_worker(rank, world, port):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank),
        WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
    )
    _patch_gpu_probes()
    import torch.nn as nn

    from stoke import DDPConfig, HorovodConfig, Stoke, StokeOptimizer

    torch.manual_seed(20 + rank)
    s = Stoke(
        model=nn.Linear(8, 4),
        optimizer=StokeOptimizer(
            optimizer=torch.optim.SGD, optimizer_kwargs={"lr": 0.1}
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        gpu=True,
        distributed="horovod",
        configs=[
            DDPConfig(local_rank=rank, backend="gloo"),
            HorovodConfig(gradient_predivide_factor=2.0, op="Sum"),
        ],
        verbose=False,
    )
    assert s.is_horovod
    eng = s._runner._engine
    # Horovod knobs mapped onto the DDP engine: pre-divide + Sum (no average)
    assert eng._predivide == 2.0 and eng._average is False
    x = torch.randn(4, 8)
    y = torch.randint(0, 4, (4,))
    out = s.model(x)
    s.backward(s.loss(out, y))
    s.step()
    # grads reduced: params equal across ranks after the step
    flat = torch.cat([p.detach().reshape(-1) for p in s.model_access.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref, atol=1e-6)
    dist.destroy_process_group()


def test_facade_horovod_compat_gloo():
    torch.multiprocessing.spawn(
        _hvd_worker, args=(2, free_port()), nprocs=2, join=True
    )


def _ds_worker(rank, world, port):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank),
        WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
    )
    _patch_gpu_probes()
    import torch.nn as nn

    from stoke import (
        DeepspeedConfig,
        DeepspeedFP16Config,
        DeepspeedZeROConfig,
        Stoke,
        StokeOptimizer,
    )

    torch.manual_seed(30 + rank)
    s = Stoke(
        model=nn.Linear(8, 4),
        optimizer=StokeOptimizer(
            optimizer=torch.optim.AdamW, optimizer_kwargs={"lr": 1e-2}
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        gpu=True,
        fp16="deepspeed",
        distributed="deepspeed",
        configs=[DeepspeedConfig(
            dist_backend="gloo",
            zero_optimization=DeepspeedZeROConfig(stage=1),
            fp16=DeepspeedFP16Config(),
        )],
        verbose=False,
    )
    # ZeRO-1 maps onto the in-house OSS engine; ds fp16 onto the native scaler
    from stoke.shard import OSSOptimizer

    assert s._runner._shard == "oss"
    assert isinstance(s.optimizer, OSSOptimizer)
    assert s.scaler is not None
    x = torch.randn(4, 8)
    y = torch.randint(0, 4, (4,))
    for _ in range(2):  # deepspeed contract: step called every micro-batch
        out = s.model(x)
        s.backward(s.loss(out, y))
        s.step()
    flat = torch.cat([p.detach().reshape(-1) for p in s.model_access.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref, atol=1e-6)
    dist.destroy_process_group()


def test_facade_deepspeed_zero1_gloo():
    torch.multiprocessing.spawn(
        _ds_worker, args=(2, free_port()), nprocs=2, join=True
    )


def _ds_nvme_worker(rank, world, port, tmpdir):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank),
        WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
    )
    _patch_gpu_probes()
    import torch.nn as nn

    from stoke import (
        DeepspeedConfig,
        DeepspeedFP16Config,
        DeepspeedOffloadOptimizerConfig,
        DeepspeedZeROConfig,
        Stoke,
        StokeOptimizer,
    )
    from stoke.ops.fused_adam import FusedAdamW

    torch.manual_seed(77)
    s = Stoke(
        model=nn.Linear(8, 4),
        optimizer=StokeOptimizer(
            optimizer=FusedAdamW, optimizer_kwargs={"lr": 1e-2}
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        gpu=True,
        fp16="deepspeed",
        distributed="deepspeed",
        configs=[DeepspeedConfig(
            dist_backend="gloo",
            fp16=DeepspeedFP16Config(),
            zero_optimization=DeepspeedZeROConfig(
                stage=1,
                offload_optimizer=DeepspeedOffloadOptimizerConfig(
                    device="nvme", nvme_path=tmpdir),
            ),
        )],
        verbose=False,
    )
    x = torch.randn(4, 8)
    y = torch.randint(0, 4, (4,))
    for _ in range(3):
        out = s.model(x)
        s.backward(s.loss(out, y))
        s.step()
    # state landed in files under nvme_path
    files = [f for f in os.listdir(tmpdir) if f.startswith("adamw_state_")]
    assert files, "NVMe-tier state files missing"
    dist.destroy_process_group()


def test_facade_deepspeed_nvme_offload_gloo(tmp_path):
    torch.multiprocessing.spawn(
        _ds_nvme_worker, args=(1, free_port(), str(tmp_path)), nprocs=1,
        join=True,
    )
