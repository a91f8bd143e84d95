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


class FlopsProfiler:
    """Per-module FLOPs / parameter profiler (the in-house equivalent of the
    DeepSpeed flops profiler the reference exposed via ``DeepspeedFlopsConfig``,
    reference ``configs.py:251-279`` -> ``distributed.py:985-1004``).

    Counts multiply-accumulates as 2 FLOPs via forward hooks on leaf modules
    (Linear / Conv2d / BatchNorm-like / activations); functional ops that do
    not pass through a module (e.g. ``F.scaled_dot_product_attention``) are
    not seen — the printed total is a floor, like most hook-based profilers.
    """

    def __init__(self, model: torch.nn.Module):
        self.model = model
        self._handles: List = []
        self.flops: Dict[str, float] = {}
        self.params: Dict[str, int] = {}

    @staticmethod
    def _module_flops(mod: torch.nn.Module, inp, out) -> float:
        if isinstance(mod, torch.nn.Linear):
            return 2.0 * out.numel() * mod.in_features
        if isinstance(mod, torch.nn.Conv2d):
            k = mod.kernel_size[0] * mod.kernel_size[1]
            cin = mod.in_channels // mod.groups
            return 2.0 * out.numel() * cin * k
        if isinstance(mod, (torch.nn.BatchNorm2d, torch.nn.LayerNorm)):
            return 4.0 * out.numel()
        if mod.__class__.__name__ in ("FusedBNAct2d", "StokeRMSNorm"):
            return 4.0 * out.numel()
        if isinstance(mod, (torch.nn.ReLU, torch.nn.GELU, torch.nn.SiLU)):
            return float(out.numel())
        if isinstance(mod, torch.nn.Embedding):
            return 0.0
        return 0.0

    def start_profile(self):
        self.flops.clear()
        self.params.clear()

        def make_hook(name):
            def hook(mod, inp, out):
                o = out[0] if isinstance(out, (list, tuple)) else out
                if isinstance(o, torch.Tensor):
                    self.flops[name] = self.flops.get(name, 0.0) + \
                        self._module_flops(mod, inp, o)
            return hook

        for name, mod in self.model.named_modules():
            if len(list(mod.children())) == 0:  # leaves only
                self.params[name] = sum(p.numel() for p in mod.parameters())
                self._handles.append(mod.register_forward_hook(make_hook(name)))

    def reset_flops(self):
        """Zero the FLOP counters while keeping hooks attached — called at
        the step boundary before the profiled step so the printed numbers
        cover exactly ONE optimizer step (accumulating
        from construction inflated the profile by the number of forwards
        under profile_step > 1 / gradient accumulation)."""
        self.flops.clear()

    def stop_profile(self):
        for h in self._handles:
            h.remove()
        self._handles.clear()

    def get_total_flops(self) -> float:
        return sum(self.flops.values())

    def get_total_params(self) -> int:
        return sum(self.params.values())

    def print_model_profile(self, top_modules: int = 1, detailed: bool = True,
                            output_file: Optional[str] = None):
        lines = [
            f"FLOPs profile: total fwd flops {self.get_total_flops()/1e9:.2f} GFLOPs, "
            f"params {self.get_total_params()/1e6:.2f} M"
        ]
        if detailed:
            ranked = sorted(self.flops.items(), key=lambda kv: -kv[1])
            for name, fl in ranked[: max(top_modules, 1) * 10]:
                lines.append(
                    f"  {name:60s} {fl/1e9:10.3f} GFLOPs "
                    f"{self.params.get(name, 0)/1e6:8.2f} M params"
                )
        text = "\n".join(lines)
        if output_file:
            with open(output_file, "w") as f:
                f.write(text + "\n")
        else:
            unrolled_print(lines)
        return text


class LazyLoss:
    """Float-like handle on an in-flight loss all-reduce.

    ``sync_loss`` launches the scalar all-reduce asynchronously and returns
    one of these instead of blocking on ``.item()`` every micro-batch
    (the reference D2H-synced + barriered per
    micro-batch at ``distributed.py:619-646``).  Arithmetic (the facade's
    agg/EMA bookkeeping) stays device-side and lazy; the single host sync
    happens only when a float is actually needed (print helpers,
    ``float()``, ``format``).
    """

    __slots__ = ("_t", "_work", "_div")

    def __init__(self, t, work=None, div=1.0):
        self._t = t
        self._work = work
        self._div = float(div)

    # -------------------------------------------------------------- plumbing
    def _tensor(self):
        if self._work is not None:
            self._work.wait()
            self._work = None
        if self._div != 1.0:
            self._t = self._t / self._div
            self._div = 1.0
        return self._t

    @staticmethod
    def _raw(x):
        return x._tensor() if isinstance(x, LazyLoss) else x

    def item(self) -> float:
        return float(self._tensor().item())

    # ------------------------------------------------------------ float-like
    def __float__(self):
        return self.item()

    def __format__(self, spec):
        return format(self.item(), spec)

    def __repr__(self):
        return repr(self.item())

    def __add__(self, other):
        return LazyLoss(self._tensor() + self._raw(other))

    __radd__ = __add__

    def __sub__(self, other):
        return LazyLoss(self._tensor() - self._raw(other))

    def __rsub__(self, other):
        return LazyLoss(self._raw(other) - self._tensor())

    def __mul__(self, other):
        return LazyLoss(self._tensor() * self._raw(other))

    __rmul__ = __mul__

    def __truediv__(self, other):
        return LazyLoss(self._tensor() / self._raw(other))

    def __rtruediv__(self, other):
        return LazyLoss(self._raw(other) / self._tensor())

    def __neg__(self):
        return LazyLoss(-self._tensor())

    def __abs__(self):
        return LazyLoss(self._tensor().abs())

    def __round__(self, ndigits=None):
        return round(self.item(), ndigits)

    def __lt__(self, other):
        return self.item() < float(self._raw(other))

    def __le__(self, other):
        return self.item() <= float(self._raw(other))

    def __gt__(self, other):
        return self.item() > float(self._raw(other))

    def __ge__(self, other):
        return self.item() >= float(self._raw(other))

    def __eq__(self, other):
        try:
            return self.item() == float(self._raw(other))
        except (TypeError, ValueError):
            return NotImplemented
