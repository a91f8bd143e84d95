# -*- coding: utf-8 -*-
"""Fused large-vocab cross-entropy (csrc/fused_ce.hip).

The eager LM-loss path casts logits to fp32 (2x bytes of a [B*S, 50k]
tensor) and runs separate softmax fwd/bwd + NLL kernels — ~6% of a GPT-2
step plus the cast traffic.  The fused kernel computes logsumexp in one
online bf16 pass and the backward writes bf16 (softmax - onehot) * scale
directly; fp32 never materializes.
"""

import torch
import torch.nn.functional as F


class _FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, ignore_index):
        from stoke import _C

        loss_sum, n_valid, lse = _C.ce_fwd(logits, target, ignore_index)
        nv = n_valid.clamp_min(1).to(torch.float32)
        loss = (loss_sum / nv).reshape(())
        ctx.save_for_backward(logits, target, lse, nv)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, gout):
        from stoke import _C

        logits, target, lse, nv = ctx.saved_tensors
        gscale = gout.detach().reshape(1).to(torch.float32) / nv
        dlogits = _C.ce_bwd(logits, target, lse, gscale, ctx.ignore_index)
        return dlogits, None, None


def fused_cross_entropy(logits: torch.Tensor, target: torch.Tensor,
                        ignore_index: int = -100) -> torch.Tensor:
    """Mean cross-entropy over non-ignored positions (torch semantics).

    logits: [..., V]; target: [...] int64.  bf16 CUDA logits run the fused
    HIP kernels; anything else falls back to ``F.cross_entropy`` on fp32.
    """
    flat = logits.reshape(-1, logits.shape[-1])
    tgt = target.reshape(-1)
    if flat.is_cuda and flat.dtype == torch.bfloat16:
        from stoke import ops

        if ops.has_ext():
            return _FusedCEFn.apply(flat.contiguous(), tgt.contiguous(),
                                    ignore_index)
    return F.cross_entropy(flat.float(), tgt, ignore_index=ignore_index)
