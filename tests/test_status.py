# -*- coding: utf-8 -*-
"""StokeStatus rule-matrix tests (mirrors reference status.py:192-289 checks)."""

import pytest
import torch

from stoke.configs import (
    ClipGradConfig,
    ClipGradNormConfig,
    DeepspeedConfig,
    DeepspeedFP16Config,
    DeepspeedZeROConfig,
    DDPConfig,
)
from stoke.status import StokeStatus, _MissingLocalRankException


def mk(monkeypatch=None, cuda=True, **kw):
    if monkeypatch is not None:
        monkeypatch.setattr(torch.cuda, "is_available", lambda: cuda)
    defaults = dict(
        batch_size_per_device=4,
        grad_accum=1,
        grad_clip=None,
        gpu=False,
        fp16=None,
        distributed=None,
        fairscale_oss=False,
        fairscale_sddp=False,
        fairscale_fsdp=False,
        configs=None,
    )
    defaults.update(kw)
    return StokeStatus(**defaults)


def test_cpu_default_ok():
    s = mk()
    assert s.batch_size == 4 and s.grad_accum == 1
    assert s.distributed is None and not s.is_fairscale


def test_gpu_without_cuda_raises(monkeypatch):
    monkeypatch.setattr(torch.cuda, "is_available", lambda: False)
    with pytest.raises(ValueError, match="CUDA is not available"):
        mk(gpu=True)


def test_fp16_silently_dropped_without_cuda(monkeypatch):
    # Reference behavior (status.py:291-319): fp16 falls back to None on CPU
    monkeypatch.setattr(torch.cuda, "is_available", lambda: False)
    s = mk(fp16="amp")
    assert s.fp16 is None


def test_distributed_requires_gpu(monkeypatch):
    with pytest.raises(ValueError, match="Distributed requires"):
        mk(monkeypatch, cuda=True, gpu=False, distributed="ddp")


def test_fairscale_requires_ddp(monkeypatch):
    with pytest.raises(ValueError, match="Fairscale extensions"):
        mk(monkeypatch, cuda=True, gpu=True, distributed=None, fairscale_oss=True)


def test_sddp_requires_oss(monkeypatch):
    with pytest.raises(ValueError, match="SDDP requires OSS"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="ddp", fairscale_sddp=True)


def test_fsdp_standalone(monkeypatch):
    with pytest.raises(ValueError, match="FSDP does not require"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="ddp",
           fairscale_oss=True, fairscale_fsdp=True)


def test_fairscale_and_deepspeed_exclusive(monkeypatch):
    with pytest.raises(ValueError, match="Cannot use both"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="deepspeed",
           fairscale_oss=True)


def test_apex_with_fairscale_raises(monkeypatch):
    with pytest.raises(ValueError, match="does not currently support APEX"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="ddp",
           fairscale_oss=True, fp16="apex_O1")


def test_oss_clip_value_raises(monkeypatch):
    with pytest.raises(ValueError, match="clip_grad_value"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="ddp",
           fairscale_oss=True, grad_clip=ClipGradConfig(clip_value=1.0))


def test_deepspeed_fp16_requires_deepspeed_dist(monkeypatch):
    with pytest.raises(ValueError, match="requires the use of"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="ddp", fp16="deepspeed")


def test_deepspeed_dist_rejects_other_fp16(monkeypatch):
    with pytest.raises(ValueError, match="only"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="deepspeed", fp16="amp")


def test_zero_requires_deepspeed_fp16(monkeypatch):
    cfg = DeepspeedConfig(zero_optimization=DeepspeedZeROConfig(stage=2))
    with pytest.raises(ValueError, match="ZeRO"):
        mk(monkeypatch, cuda=True, gpu=True, distributed="deepspeed",
           configs=[cfg])


def test_zero_with_fp16_ok(monkeypatch):
    cfg = DeepspeedConfig(
        zero_optimization=DeepspeedZeROConfig(stage=2),
        fp16=DeepspeedFP16Config(),
    )
    s = mk(monkeypatch, cuda=True, gpu=True, distributed="deepspeed",
           fp16="deepspeed", configs=[cfg])
    assert s.zero == 2 and s.is_fp16_deepspeed


def test_grad_clip_type_check():
    with pytest.raises(TypeError):
        mk(grad_clip=1.0)


def test_effective_batch_size():
    s = mk(grad_accum=4)
    s.set_post_init_values(world_size=8)
    assert s.effective_batch_size == 4 * 4 * 8


def test_ddp_config_env_local_rank(monkeypatch):
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setenv("LOCAL_RANK", "3")
    s = mk(monkeypatch, cuda=True, gpu=True, distributed="ddp")
    assert s.ddp_config.local_rank == 3


def test_ddp_config_missing_local_rank(monkeypatch):
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.delenv("LOCAL_RANK", raising=False)
    s = mk(monkeypatch, cuda=True, gpu=True, distributed="ddp")
    with pytest.raises(_MissingLocalRankException):
        _ = s.ddp_config


def test_fsdp_config_mixed_precision_injection(monkeypatch):
    s = mk(monkeypatch, cuda=True, gpu=True, distributed="ddp",
           fairscale_fsdp=True, fp16="amp")
    assert s.fsdp_config.mixed_precision is True
    s2 = mk(monkeypatch, cuda=True, gpu=True, distributed="ddp",
            fairscale_fsdp=True)
    assert s2.fsdp_config.mixed_precision is False


def test_repr_contains_state():
    s = mk(grad_clip=ClipGradNormConfig(max_norm=1.0, norm_type=2.0))
    r = repr(s)
    assert "STOKE STATE" in r and "max_norm" in r
