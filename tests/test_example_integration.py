# -*- coding: utf-8 -*-
"""Example-as-integration: run examples/cifar10/train.py under the CPU config
(the reference's de-facto test strategy)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXAMPLE = os.path.join(REPO, "examples", "cifar10")


def _run(config, extra=()):
    return subprocess.run(
        [sys.executable, os.path.join(EXAMPLE, "train.py"),
         "--config", os.path.join(EXAMPLE, "config", config),
         "--max-steps", "4", "--batch", "8", *extra],
        capture_output=True, text=True, timeout=600,
    )


def test_example_cpu_config_runs():
    r = _run("cpu.yaml")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "loss" in (r.stdout + r.stderr).lower()


@pytest.mark.gpu
def test_example_gpu_bf16_config_runs():
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    r = _run("gpu_bf16.yaml")
    assert r.returncode == 0, r.stderr[-2000:]


def test_llm_example_cpu_runs():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "llm", "train.py"),
         "--cpu", "--layers", "2", "--dim", "256", "--steps", "3"],
        capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert "done: 3 steps" in r.stdout
