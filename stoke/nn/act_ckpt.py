# -*- coding: utf-8 -*-
"""Activation checkpointing (recompute-in-backward).

The reference exposes activation checkpointing only as a DeepSpeed
pass-through (``DeepspeedActivationCheckpointingConfig``,
reference ``stoke/configs.py:222-249`` -> ``distributed.py:965-983``); the
work itself happened inside the DeepSpeed engine.  Here it is in-house:
eligible submodules get their forward routed through
``torch.utils.checkpoint`` (non-reentrant), trading one extra forward of the
wrapped block for not storing its activations — on MI355X the recompute is
usually HBM-resident and cheap relative to the 288 GB it frees for larger
batches/models.

Composes with the FSDP engine: the recompute re-fires the unit's
forward-pre hook, which re-gathers the flat-param shard it needs.
"""

import functools
from typing import Callable, Optional

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint


_MARK = "_stoke_act_ckpt"


def _wrap_module(m: nn.Module) -> None:
    if getattr(m, _MARK, False):
        return
    inner_forward = m.forward

    @functools.wraps(inner_forward)
    def ckpt_forward(*args, **kwargs):
        if not torch.is_grad_enabled() or not m.training:
            return inner_forward(*args, **kwargs)
        return checkpoint(
            inner_forward, *args, use_reentrant=False, **kwargs
        )

    m.forward = ckpt_forward
    setattr(m, _MARK, True)


def apply_activation_checkpointing(
    model: nn.Module,
    predicate: Optional[Callable[[str, nn.Module], bool]] = None,
    min_params: int = 1_000_000,
) -> int:
    """Wrap eligible submodules with recompute-in-backward.

    Default eligibility: any non-container child subtree holding at least
    ``min_params`` parameters (the same granularity the FSDP engine uses for
    shard units).  A custom ``predicate(name, module)`` overrides it.
    Returns the number of modules wrapped.
    """
    wrapped = 0
    visited = set()

    def consider(name: str, m: nn.Module) -> bool:
        nonlocal wrapped
        if predicate is not None:
            if predicate(name, m):
                _wrap_module(m)
                wrapped += 1
                return True
            return False
        if isinstance(m, (nn.ModuleList, nn.Sequential, nn.ModuleDict)):
            return False
        n = sum(p.numel() for p in m.parameters())
        if n >= min_params:
            _wrap_module(m)
            wrapped += 1
            return True
        return False

    def recurse(root: nn.Module, prefix: str):
        for cname, child in root.named_children():
            path = f"{prefix}{cname}"
            if id(child) in visited:
                continue
            visited.add(id(child))
            if not consider(path, child):
                recurse(child, path + ".")

    recurse(model, "")
    return wrapped
