# -*- coding: utf-8 -*-
"""In-house CDNA4 flash attention (MFMA bf16, FA-2 style).

``flash_attention(q, k, v, causal=...)`` runs the hand-written kernels in
``csrc/fa_fwd.hip`` / ``csrc/fa_bwd.hip`` — hardware-validated round 2
(tests/test_fa.py: fragment-layout probe + fwd/bwd numerics vs fp32
SDPA, all passing on MI355X).  ``attention()`` is the model-facing entry:
it routes to the in-house kernels when the shapes qualify and the
``STOKE_USE_FA`` gate allows, else falls back to
``F.scaled_dot_product_attention`` (AOTriton).
"""

import os

import torch
import torch.nn.functional as F


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
        dq, dk, dv = _C.fa_bwd(q, k, v, out, dout, lse, ctx.causal)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True) -> torch.Tensor:
    """q: [B,H,S,D], k/v: [B,Hkv,S,D] bf16, D in {64, 128}.

    Any strides with a unit head dim are handled natively (the kernels
    take element strides), so transposed projection views — [B,S,H,D]
    storage — run with ZERO copies; the output is a [B,H,S,D] view of
    [B,S,H,D] storage, making the caller's transpose+reshape free too.
    """
    return _FlashAttnFn.apply(q, k, v, causal)


def _fa_usable(q, k, v, causal) -> bool:
    # Default ON: the strip-paired kernels measure 1.64-1.66x AOTriton fwd+bwd
    # on both bench shapes (D=64 GPT-2 and D=128 GQA Llama —
    # benchmarks/fa_bench.py, NOTES.md).  STOKE_USE_FA=0 disables.
    if os.environ.get("STOKE_USE_FA", "1") == "0":
        return False
    if not (q.is_cuda and q.dtype == torch.bfloat16
            and k.dtype == torch.bfloat16 and v.dtype == torch.bfloat16):
        return False
    if q.dim() != 4 or q.shape[-1] not in (64, 128):
        return False
    # self-attention only: the kernels assume one shared sequence length
    if k.shape[2] != q.shape[2] or v.shape[2] != q.shape[2]:
        return False
    if k.shape[1] != q.shape[1] and q.shape[1] % k.shape[1] != 0:
        return False
    from stoke import ops

    return ops.has_ext()


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = True) -> torch.Tensor:
    """Model-facing attention: in-house MFMA flash kernels when usable
    (bf16 CUDA [B,H,S,D] with D in {64,128}), SDPA otherwise.  Disable the
    native path with ``STOKE_USE_FA=0``."""
    if _fa_usable(q, k, v, causal):
        return flash_attention(q, k, v, causal=causal)
    return F.scaled_dot_product_attention(
        q, k, v, is_causal=causal, enable_gqa=(k.shape[1] != q.shape[1])
    )
