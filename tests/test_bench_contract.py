# -*- coding: utf-8 -*-
"""Pins the driver contract of bench.py: self-launching N ranks, one JSON
line from rank 0 with the required fields."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run_bench(args, timeout=420):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), *args],
        capture_output=True, text=True, timeout=timeout, cwd=REPO,
    )
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, f"no JSON line\nstdout:{out.stdout[-2000:]}\nstderr:{out.stderr[-2000:]}"
    return json.loads(lines[-1])


def test_bench_cpu_single():
    j = _run_bench(["--cpu", "--steps", "2", "--warmup", "1", "--batch", "4"])
    assert REQUIRED <= set(j)
    assert j["data"] == "synthetic"
    assert j["config"]["parallelism"] == "cpu"


@pytest.mark.timeout(600)
def test_bench_cpu_self_launches_two_ranks():
    """--gpus 2 with no WORLD_SIZE must exec torchrun and report n_gpus=2."""
    env = {k: v for k, v in os.environ.items()
           if k not in ("WORLD_SIZE", "RANK", "LOCAL_RANK")}
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cpu", "--gpus",
         "2", "--steps", "2", "--warmup", "1", "--batch", "4"],
        capture_output=True, text=True, timeout=480, cwd=REPO, env=env,
    )
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, out.stdout[-2000:] + out.stderr[-2000:]
    j = json.loads(lines[-1])
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert REQUIRED <= set(j)
