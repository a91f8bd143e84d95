# -*- coding: utf-8 -*-
"""In-house flash attention — ROUND-2 WORK IN PROGRESS.

``flash_attention(q, k, v, causal=...)`` runs the MFMA kernels in
``csrc/fa_fwd.hip`` / ``csrc/fa_bwd.hip``.  UNVALIDATED on hardware as of
round 1 (GPU budget exhausted after compile verification): nothing imports
this module by default, and ``benchmarks/models.py`` keeps
``F.scaled_dot_product_attention``.  Round 2: run the env-gated tests in
tests/test_fa_wip.py (probe first), fix what they find, then gate this in
via ``STOKE_USE_FA=1``.
"""

import torch


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        from stoke import _C

        out, lse = _C.fa_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, dout):
        from stoke import _C

        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = _C.fa_bwd(q, k, v, out, dout.contiguous(), lse,
                               ctx.causal)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True) -> torch.Tensor:
    """q: [B,H,S,D], k/v: [B,Hkv,S,D] bf16 contiguous; D in {64, 128}."""
    return _FlashAttnFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                              causal)
