# -*- coding: utf-8 -*-
"""Python bindings for the native HIP kernels (csrc/stoke_kernels.hip).

Dispatch policy:
* CUDA/ROCm tensors -> the in-tree gfx950 extension ``stoke._C``.  If the
  extension is missing on a GPU machine these functions raise loudly instead
  of silently falling back to eager torch (so a GPU run always exercises the
  native path).
* CPU tensors -> plain torch reference implementations (used by the CPU test
  tier; also the numerics reference the GPU kernels are tested against).
"""

from typing import List, Optional

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from stoke import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
    except ImportError as e:  # remember why so the error message is useful
        _EXT_ERR = str(e)
    return _EXT


def has_ext() -> bool:
    return _load_ext() is not None


def _require_ext():
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "stoke -- native HIP extension 'stoke._C' is not built but a GPU "
            "tensor was passed. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_EXT_ERR}"
        )
    return ext


# ---------------------------------------------------------------------------
# Functional API
# ---------------------------------------------------------------------------
def multi_tensor_unscale_(
    grads: List[torch.Tensor], inv_scale: torch.Tensor, found_inf: torch.Tensor
):
    """grads *= inv_scale, setting found_inf=1.0 on any non-finite value."""
    if not grads:
        return
    if grads[0].is_cuda:
        _require_ext().multi_tensor_unscale_(grads, inv_scale, found_inf)
    else:
        inv = inv_scale.item()
        for g in grads:
            g.mul_(inv)
            if not torch.isfinite(g).all():
                found_inf.fill_(1.0)


def multi_tensor_l2norm(tensors: List[torch.Tensor]) -> torch.Tensor:
    """Global L2 norm over a tensor list; returns a 1-element device tensor."""
    if not tensors:
        return torch.zeros(1)
    if tensors[0].is_cuda:
        sq = _require_ext().multi_tensor_l2norm_sq(tensors)
        return sq.sqrt()
    sq = sum(t.float().pow(2).sum() for t in tensors)
    return sq.sqrt().reshape(1)


def multi_tensor_scale_(tensors: List[torch.Tensor], scale: torch.Tensor):
    """In-place multiply every tensor by a device-scalar (0-dim/1-elem) tensor."""
    if not tensors:
        return
    if tensors[0].is_cuda:
        _require_ext().multi_tensor_scale_(tensors, scale)
    else:
        s = scale.item()
        for t in tensors:
            t.mul_(s)


def multi_tensor_clamp_(tensors: List[torch.Tensor], limit: float):
    """In-place clamp every tensor to [-limit, limit]."""
    if not tensors:
        return
    if tensors[0].is_cuda:
        _require_ext().multi_tensor_clamp_(tensors, float(limit))
    else:
        for t in tensors:
            t.clamp_(-limit, limit)


def fused_adamw_(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    exp_avgs: List[torch.Tensor],
    exp_avg_sqs: List[torch.Tensor],
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    found_inf: Optional[torch.Tensor] = None,
    inv_scale: Optional[torch.Tensor] = None,
    masters: Optional[List[torch.Tensor]] = None,
):
    """Single-launch fused AdamW over the whole parameter list.

    fp32 path: params/grads/state all fp32.  bf16 path (``masters`` given):
    bf16 params+grads with fp32 master weights and state (FSDP flat shards).
    """
    if not params:
        return
    if params[0].is_cuda:
        ext = _require_ext()
        if masters is not None:
            ext.multi_tensor_adamw_bf16_(
                params, grads, exp_avgs, exp_avg_sqs, masters,
                step, lr, beta1, beta2, eps, weight_decay, found_inf, inv_scale,
            )
        else:
            ext.multi_tensor_adamw_(
                params, grads, exp_avgs, exp_avg_sqs,
                step, lr, beta1, beta2, eps, weight_decay, found_inf, inv_scale,
            )
        return
    # CPU reference implementation (also the numerics oracle for GPU tests)
    if found_inf is not None and found_inf.item() != 0:
        return
    inv = inv_scale.item() if inv_scale is not None else 1.0
    bc1 = 1.0 - beta1**step
    bc2 = 1.0 - beta2**step
    for i, p in enumerate(params):
        g = grads[i].float() * inv
        m, v = exp_avgs[i], exp_avg_sqs[i]
        w = masters[i] if masters is not None else p
        m.mul_(beta1).add_(g, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
        denom = (v / bc2).sqrt().add_(eps)
        update = (m / bc1) / denom + weight_decay * w.float()
        w.add_(update, alpha=-lr)
        if masters is not None:
            p.copy_(w.to(p.dtype))


def fused_sgd_(
    params: List[torch.Tensor],
    grads: List[torch.Tensor],
    momentum_bufs: List[torch.Tensor],
    lr: float,
    momentum: float = 0.0,
    dampening: float = 0.0,
    weight_decay: float = 0.0,
    nesterov: bool = False,
    first_step: bool = False,
    found_inf: Optional[torch.Tensor] = None,
    inv_scale: Optional[torch.Tensor] = None,
    masters: Optional[List[torch.Tensor]] = None,
):
    """Single-launch fused SGD(+momentum) over the whole parameter list.

    ``torch.optim.SGD`` semantics including the first-step ``buf = g``
    initialization.  fp32 path, or bf16 params+grads with fp32
    masters+buffers when ``masters`` is given.
    """
    if not params:
        return
    if params[0].is_cuda:
        ext = _require_ext()
        if masters is not None:
            ext.multi_tensor_sgd_bf16_(
                params, grads, momentum_bufs, masters, lr, momentum,
                dampening, weight_decay, nesterov, first_step,
                found_inf, inv_scale,
            )
        else:
            ext.multi_tensor_sgd_(
                params, grads, momentum_bufs, lr, momentum, dampening,
                weight_decay, nesterov, first_step, found_inf, inv_scale,
            )
        return
    # CPU reference implementation (the numerics oracle for GPU tests)
    if found_inf is not None and found_inf.item() != 0:
        return
    inv = inv_scale.item() if inv_scale is not None else 1.0
    for i, p in enumerate(params):
        w = masters[i] if masters is not None else p
        g = grads[i].float() * inv + weight_decay * w.float()
        if momentum != 0.0:
            b = momentum_bufs[i]
            if first_step:
                b.copy_(g)
            else:
                b.mul_(momentum).add_(g, alpha=1 - dampening)
            g = g + momentum * b if nesterov else b.clone()
        w.add_(g, alpha=-lr)
        if masters is not None:
            p.copy_(w.to(p.dtype))


def amp_update_scale_(
    scale: torch.Tensor,
    growth_tracker: torch.Tensor,
    found_inf: torch.Tensor,
    growth_factor: float,
    backoff_factor: float,
    growth_interval: int,
):
    """Dynamic loss-scale update (device-side on GPU, host math on CPU)."""
    if scale.is_cuda:
        _require_ext().amp_update_scale_(
            scale, growth_tracker, found_inf,
            growth_factor, backoff_factor, growth_interval,
        )
        return
    if found_inf.item() != 0:
        scale.mul_(backoff_factor)
        growth_tracker.fill_(0)
    else:
        g = int(growth_tracker.item()) + 1
        if g >= growth_interval:
            scale.mul_(growth_factor)
            g = 0
        growth_tracker.fill_(g)
