# -*- coding: utf-8 -*-
"""Fused LayerNorm module (csrc/fused_layernorm.hip): one streaming kernel
each way on bf16 GPU inputs, eager ``F.layer_norm`` elsewhere.  State-dict
compatible with ``torch.nn.LayerNorm`` (weight, bias)."""

import torch
import torch.nn as nn
import torch.nn.functional as F


def _can_fuse(x: torch.Tensor, w: torch.Tensor) -> bool:
    from stoke import ops

    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and w.dtype == torch.bfloat16
        and x.shape[-1] % 8 == 0
        and x.shape[-1] <= 16384
        and x.is_contiguous()
        and ops.has_ext()
    )


class _LNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        from stoke import _C

        y, mean, invstd = _C.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        from stoke import _C

        x, weight, mean, invstd = ctx.saved_tensors
        if not dy.is_contiguous():
            dy = dy.contiguous()
        dx, dw, db = _C.layernorm_bwd(x, dy, weight, mean, invstd)
        return dx, dw, db, None


class StokeLayerNorm(nn.LayerNorm):
    """Drop-in nn.LayerNorm (single normalized dim) with a fused HIP path."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (
            len(self.normalized_shape) == 1
            and self.bias is not None
            and _can_fuse(x, self.weight)
            and self.bias.dtype == torch.bfloat16
        ):
            return _LNFn.apply(x, self.weight, self.bias, self.eps)
        return F.layer_norm(
            x, self.normalized_shape, self.weight, self.bias, self.eps
        )
