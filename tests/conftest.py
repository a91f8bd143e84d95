# -*- coding: utf-8 -*-
import os
import socket

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run on MI355X box)"
    )


def free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def dist_env(monkeypatch):
    """Env for single-process 'distributed' tests."""
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", str(free_port()))
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    monkeypatch.setenv("LOCAL_RANK", "0")


def run_spawn(fn, world_size=2, args=()):
    """Spawn world_size processes running fn(rank, world_size, port, *args)."""
    port = free_port()
    torch.multiprocessing.spawn(
        fn, args=(world_size, port) + tuple(args), nprocs=world_size, join=True
    )


def init_gloo(rank: int, world_size: int, port: int):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    from stoke.comm import StokeProcessGroup

    return StokeProcessGroup(backend="gloo", init_method="env://", local_rank=rank)
