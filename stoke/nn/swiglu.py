# -*- coding: utf-8 -*-
"""Fused SwiGLU: y = silu(gate) * up, one streaming kernel each way
(csrc/fused_swiglu.hip).  Eager fallback on CPU / non-bf16 inputs."""

import torch
import torch.nn.functional as F


def _can_fuse(g: torch.Tensor, u: torch.Tensor) -> bool:
    from stoke import ops

    return (
        g.is_cuda
        and g.dtype == torch.bfloat16
        and u.dtype == torch.bfloat16
        and g.is_contiguous()
        and u.is_contiguous()
        and g.numel() % 8 == 0
        and ops.has_ext()
    )


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, u):
        from stoke import _C

        ctx.save_for_backward(g, u)
        return _C.swiglu_fwd(g, u)

    @staticmethod
    def backward(ctx, dy):
        from stoke import _C

        g, u = ctx.saved_tensors
        if not dy.is_contiguous():
            dy = dy.contiguous()
        dg, du = _C.swiglu_bwd(dy, g, u)
        return dg, du


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up."""
    if _can_fuse(gate, up):
        return _SwiGLUFn.apply(gate, up)
    return F.silu(gate) * up
