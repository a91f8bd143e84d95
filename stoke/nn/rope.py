# -*- coding: utf-8 -*-
"""Fused rotary position embedding (interleaved-pair RoPE).

``apply_rope(x, cos, sin)`` rotates a CONTIGUOUS [B, S, H, Dh] bf16
projection in one HIP kernel (csrc/fused_rope.hip); the backward is the same
kernel with the rotation conjugated.  Falls back to the eager strided
composition on CPU / non-bf16 inputs.
"""

import torch


def _can_fuse(x: torch.Tensor) -> bool:
    from stoke import ops

    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.dim() == 4
        and x.is_contiguous()
        and x.shape[-1] % 8 == 0
        and ops.has_ext()
    )


def _eager_rope(x, cos, sin, conj=False):
    # x: [..., S, H, Dh] or [B, H, S, D]-agnostic as long as cos broadcast
    # matches dim -3 (handled by callers); here x is [B, S, H, Dh]
    if conj:
        sin = -sin
    x1, x2 = x[..., 0::2], x[..., 1::2]
    S = x.shape[1]
    c = cos[:S][None, :, None, :]
    s = sin[:S][None, :, None, :]
    out = torch.empty_like(x)
    out[..., 0::2] = x1 * c - x2 * s
    out[..., 1::2] = x2 * c + x1 * s
    return out


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        from stoke import _C

        ctx.save_for_backward(cos, sin)
        return _C.rope_apply(x, cos, sin, False)

    @staticmethod
    def backward(ctx, dy):
        from stoke import _C

        cos, sin = ctx.saved_tensors
        if not dy.is_contiguous():
            dy = dy.contiguous()
        return _C.rope_apply(dy, cos, sin, True), None, None


def apply_rope(x: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> torch.Tensor:
    """Rotate pairs (x[2i], x[2i+1]) of the last dim by position-dependent
    angles.  x: [B, S, H, Dh]; cos/sin: [>=S, Dh/2] fp32."""
    if _can_fuse(x):
        return _RopeFn.apply(x, cos.contiguous(), sin.contiguous())
    return _eager_rope(x, cos, sin)
