# -*- coding: utf-8 -*-
"""FSDP overlap ORDER logic on CPU: mock streams/events let the prefetch
bookkeeping (first-forward order recording, depth-1 forward/backward
prefetch, event fencing flags) run without a GPU."""

import os

import pytest
import torch
import torch.nn as nn

from tests.conftest import free_port


class _FakeStream:
    def wait_stream(self, other):
        pass


class _FakeEvent:
    def __init__(self):
        self.recorded = 0

    def record(self, stream=None):
        self.recorded += 1


class _FakeCurrent(_FakeStream):
    def wait_event(self, ev):
        pass


def _worker(rank, world, port):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    import torch.distributed as dist

    from stoke.comm import StokeProcessGroup
    from stoke.shard import StokeFSDPModule

    pg = StokeProcessGroup(backend="gloo", init_method="env://",
                           local_rank=rank)
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(16, 32), nn.Tanh(), nn.Linear(32, 32),
                          nn.Tanh(), nn.Linear(32, 4))
    fsdp = StokeFSDPModule(model, pg=pg, compute_dtype=torch.float32,
                           min_wrap_params=100)
    # inject mock comm-stream machinery (gloo normally disables it)
    fsdp._comm_stream = _FakeStream()
    torch.cuda.current_stream = lambda *a, **k: _FakeCurrent()
    torch.cuda.Event = _FakeEvent
    torch.Tensor.record_stream = lambda self, s: None
    import torch.cuda as tc

    class _Ctx:
        def __init__(self, s): pass
        def __enter__(self): return None
        def __exit__(self, *a): return False

    tc.stream = _Ctx
    x = torch.randn(4, 16)
    y = torch.randint(0, 4, (4,))
    for step in range(2):
        loss = nn.CrossEntropyLoss()(fsdp(x), y)
        loss.backward()
        fsdp.finish_backward()
        for u in fsdp.units:
            assert u.shard.grad is not None
            u.shard.grad = None
    # order was recorded on the first forward and prefetch used it
    wrapped = [u for u in fsdp.units if u.name != "(root)"]
    assert fsdp._order_final
    assert [u.name for u in fsdp._fwd_order] == [u.name for u in wrapped]
    # async gathers really went through the mocked comm stream
    assert any(u._gather_event is not None and u._gather_event.recorded > 0
               for u in wrapped), "no prefetch ever used the comm stream"
    dist.destroy_process_group()


def test_fsdp_prefetch_order_cpu():
    torch.multiprocessing.spawn(
        _worker, args=(1, free_port()), nprocs=1, join=True
    )
