# -*- coding: utf-8 -*-
"""Unified checkpoint save/load.

Preserves the reference checkpoint contract exactly (SURVEY.md section 5.4,
reference ``stoke/io_ops.py:49-87, 224-236``):

* file name  : ``{path}/stoke-{name}-backward-step-{backward_step}.{ext}``
* payload    : ``backward_step``, ``grad_accum_step``, ``optimizer_step``,
               ``stoke_status``, ``model_state_dict``, ``optimizer_state_dict``,
               ``scaler_state_dict``, ``extras``
* semantics  : saved on rank 0 behind collective fences; sharded state (OSS /
               SDDP / FSDP) is consolidated to full, world-size-independent
               state dicts before writing, and re-sharded on load.
"""

from typing import Callable, Optional

import torch

from stoke.utils import make_folder

_PREFIX = "stoke"


def make_tag(name: str, backward_step: int) -> str:
    return f"{_PREFIX}-{name}-backward-step-{backward_step}"


def make_full_save_path(path: str, name: str, backward_step: int, extension: str) -> str:
    return f"{path}/{make_tag(name, backward_step)}.{extension}"


def _collect_state_dicts(runner, shard, model, optimizer):
    """Produce world-size-independent model/optimizer state dicts."""
    from stoke.shard import OSSOptimizer, StokeFSDPModule

    if shard == "fsdp" and isinstance(model, StokeFSDPModule):
        model_dict = model.full_state_dict()
        optimizer_dict = model.gather_full_optim_state_dict(optimizer)
    elif isinstance(optimizer, OSSOptimizer):
        model_dict = model.state_dict()
        optimizer_dict = optimizer.consolidate_state_dict(recipient_rank=0)
    else:
        model_dict = model.state_dict()
        optimizer_dict = optimizer.state_dict()
    return model_dict, optimizer_dict


def save_checkpoint(
    runner,
    shard: str,
    model: torch.nn.Module,
    optimizer,
    path: str,
    backward_step: int,
    grad_accum_step: int,
    optimizer_step: int,
    name: str,
    status: dict,
    scaler_dict: Optional[dict] = None,
    extension: str = "pt",
    create_directory: bool = True,
    extras: Optional[dict] = None,
    verbose: bool = True,
    save_rank: int = 0,
):
    save_path = make_full_save_path(path, name, backward_step, extension)
    distributed = not isinstance(runner.rank, str)
    if distributed:
        runner.barrier()
    model_dict, optimizer_dict = _collect_state_dicts(runner, shard, model, optimizer)
    is_writer = (not distributed) or runner.rank == save_rank
    if is_writer:
        if create_directory:
            make_folder(path)
        try:
            torch.save(
                {
                    "backward_step": backward_step,
                    "grad_accum_step": grad_accum_step,
                    "optimizer_step": optimizer_step,
                    "stoke_status": status,
                    "model_state_dict": model_dict,
                    "optimizer_state_dict": optimizer_dict,
                    "scaler_state_dict": scaler_dict,
                    "extras": extras,
                },
                save_path,
            )
        except OSError as e:
            print(f"Stoke -- Unable to save model to given path: {save_path}")
            raise e
    if distributed:
        runner.barrier()
    return path, f"{make_tag(name, backward_step)}.{extension}"


def load_checkpoint(
    runner,
    shard: str,
    model: torch.nn.Module,
    optimizer,
    gpu: bool,
    path: str,
    tag: str,
    scaler_dict_fn: Optional[Callable] = None,
    strict: bool = True,
):
    from stoke.shard import OSSOptimizer, StokeFSDPModule

    if gpu:
        dev = runner.device_id
        map_loc = f"cuda:{dev}" if not isinstance(dev, str) else "cuda"
    else:
        map_loc = "cpu"
    load_dict = torch.load(f"{path}/{tag}", map_location=map_loc,
                           weights_only=False)
    if shard == "fsdp" and isinstance(model, StokeFSDPModule):
        model.load_full_state_dict(load_dict["model_state_dict"], strict=strict)
        if load_dict.get("optimizer_state_dict") is not None:
            model.load_full_optim_state_dict(
                optimizer, load_dict["optimizer_state_dict"]
            )
    elif isinstance(optimizer, OSSOptimizer):
        model.load_state_dict(load_dict["model_state_dict"], strict=strict)
        if load_dict.get("optimizer_state_dict") is not None:
            optimizer.load_full_state_dict(load_dict["optimizer_state_dict"])
    else:
        model.load_state_dict(load_dict["model_state_dict"], strict=strict)
        if load_dict.get("optimizer_state_dict") is not None:
            optimizer.load_state_dict(load_dict["optimizer_state_dict"])
    if scaler_dict_fn is not None and load_dict.get("scaler_state_dict") is not None:
        scaler_dict_fn(load_dict["scaler_state_dict"])
    return (
        load_dict["backward_step"],
        load_dict["grad_accum_step"],
        load_dict["optimizer_step"],
        load_dict["extras"],
    )
