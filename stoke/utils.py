# -*- coding: utf-8 -*-
"""Utility helpers shared across the framework.

Provides the same public helpers as the reference (``stoke/utils.py:30-151``):
``ParamNormalize``, ``place_data_on_gpu``, ``zero_optimizer_grads``,
``unrolled_print`` and ``make_folder``.
"""

import os
from enum import Enum
from typing import Any, Callable, Dict, List, Optional, Tuple, TypeVar, Union

import torch

T_co = TypeVar("T_co", covariant=True)
T = TypeVar("T")

_worker_init_fn_t = Callable[[int], None]
_collate_fn_t = Callable[[List[T]], Any]


class ParamNormalize(Enum):
    """Normalization factors for pretty-printing parameter counts."""

    THOUSAND = 1e3
    MILLION = 1e6
    BILLION = 1e9
    TRILLION = 1e12


def place_data_on_gpu(
    data: Union[
        torch.Tensor,
        List[torch.Tensor],
        Tuple[torch.Tensor],
        Dict[str, torch.Tensor],
    ],
    fp16: Optional[str] = None,
    device: Optional[torch.device] = None,
):
    """Recursively move a tensor / list / tuple / dict of tensors onto the GPU.

    Matches reference semantics (``utils.py:39-80``): under the deepspeed-style
    fp16 mode inputs are cast to half, otherwise the dtype is preserved.
    Non-tensor leaves without a ``.to`` method pass through untouched.
    """
    if device is None:
        device = torch.device("cuda")
    if isinstance(data, torch.Tensor):
        if fp16 == "deepspeed" and data.is_floating_point():
            return data.to(device=device, dtype=torch.half, non_blocking=True)
        return data.to(device=device, non_blocking=True)
    elif isinstance(data, (list, tuple)):
        return type(data)(place_data_on_gpu(val, fp16, device) for val in data)
    elif isinstance(data, dict):
        return {k: place_data_on_gpu(v, fp16, device) for k, v in data.items()}
    elif not hasattr(data, "to"):
        return data
    else:
        return data.to(device=device)


def zero_optimizer_grads(
    optimizer: torch.optim.Optimizer,
    apex: bool = False,
    horovod: bool = False,
):
    """Zero gradients, choosing ``set_to_none`` when safe.

    Fused multi-tensor optimizers (our HIP FusedAdamW included, detected by
    "Fused" in the class name) keep gradients allocated between steps, so they
    are zeroed in place; everything else gets ``set_to_none=True`` (reference
    semantics, ``utils.py:83-106``).
    """
    prefers_none = getattr(optimizer, "zero_grad_prefers_none", False)
    if prefers_none or (
        optimizer.__class__.__name__.find("Fused") == -1 and not apex and not horovod
    ):
        optimizer.zero_grad(set_to_none=True)
    else:
        optimizer.zero_grad(set_to_none=False)


def unrolled_print(msg: Union[str, List[str], Tuple[str]], single_line: bool = False):
    """Print a message or iterable of messages with the "Stoke -- " prefix."""
    if isinstance(msg, (list, tuple)):
        if single_line:
            msg = type(msg)(
                f"Stoke -- {val}" if idx == 0 else f"{val}"
                for idx, val in enumerate(msg)
            )
        else:
            msg = type(msg)(f"Stoke -- {val}" for val in msg)
        print(*msg, sep=", " if single_line else "\n")
    else:
        print(f"Stoke -- {msg}")


def make_folder(path: str):
    """Create a directory (and parents) if it does not already exist."""
    if not os.path.isdir(path):
        os.makedirs(path, exist_ok=True)
