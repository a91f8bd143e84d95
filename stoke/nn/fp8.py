# -*- coding: utf-8 -*-
"""FP8 (OCP e4m3/e5m2) linear layers for CDNA4 fp8 MFMA — the stretch
precision mode past bf16 (MI355X fp8 dense peak is ~2x bf16).

``FP8Linear`` runs its GEMMs through ``torch._scaled_mm`` (hipBLASLt fp8)
with per-tensor dynamic scaling:

* forward  : x, w cast to e4m3 with amax-based scales -> fp8 GEMM -> bf16 out
* backward : dy cast to e5m2 (wider exponent for gradients);
             dx = dy @ w   and   dw = dy^T @ x   both as fp8 GEMMs

``fp8_available()`` probes the dtypes + kernel once; on machines without fp8
support the layer behaves exactly like ``nn.Linear`` (bf16/fp32 matmul), so
models stay portable.
"""

from typing import Optional

import torch
import torch.nn as nn

_E4M3 = getattr(torch, "float8_e4m3fn", None)
_E5M2 = getattr(torch, "float8_e5m2", None)

_PROBED: Optional[bool] = None


def fp8_available() -> bool:
    """True when torch._scaled_mm works with OCP fp8 dtypes on this device."""
    global _PROBED
    if _PROBED is not None:
        return _PROBED
    if _E4M3 is None or not torch.cuda.is_available():
        _PROBED = False
        return False
    try:
        a = torch.randn(16, 16, device="cuda").to(_E4M3)
        b = torch.randn(16, 16, device="cuda").to(_E4M3)
        s = torch.ones((), device="cuda")
        out = torch._scaled_mm(a, b.t(), scale_a=s, scale_b=s,
                               out_dtype=torch.bfloat16)
        _PROBED = bool(out.shape == (16, 16))
    except (RuntimeError, TypeError, AttributeError):
        _PROBED = False
    return _PROBED


def _amax_scale(t: torch.Tensor, fp8_max: float) -> torch.Tensor:
    amax = t.abs().amax().float().clamp(min=1e-12)
    return (amax / fp8_max).clamp(min=1e-12)


def _to_fp8(t: torch.Tensor, dtype, fp8_max: float):
    scale = _amax_scale(t, fp8_max)
    q = (t.float() / scale).clamp(-fp8_max, fp8_max).to(dtype)
    return q, scale


class _FP8MMFn(torch.autograd.Function):
    """y = x @ w^T in fp8; x: [T, K] bf16, w: [N, K] bf16 -> y: [T, N] bf16."""

    @staticmethod
    def forward(ctx, x, w):
        x8, sx = _to_fp8(x, _E4M3, 448.0)
        w8, sw = _to_fp8(w, _E4M3, 448.0)
        # mat_a row-major [T,K]; mat_b column-major [K,N] = view w8[N,K].t()
        y = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                             out_dtype=torch.bfloat16)
        ctx.save_for_backward(x8, sx, w8, sw)
        return y

    @staticmethod
    def backward(ctx, dy):
        x8, sx, w8, sw = ctx.saved_tensors
        dy = dy.contiguous()
        dy8, sdy = _to_fp8(dy, _E5M2, 57344.0)
        # dx[T,K] = dy[T,N] @ w[N,K]: mat_b must be column-major [N,K]
        w8_col = w8.t().contiguous().t()
        dx = torch._scaled_mm(dy8, w8_col, scale_a=sdy, scale_b=sw,
                              out_dtype=torch.bfloat16)
        # dw[N,K] = dy^T[N,T] @ x[T,K]: mat_a row-major, mat_b column-major
        x8_col = x8.t().contiguous().t()
        dw = torch._scaled_mm(dy8.t().contiguous(), x8_col, scale_a=sdy,
                              scale_b=sx, out_dtype=torch.bfloat16)
        return dx, dw


class FP8Linear(nn.Linear):
    """Drop-in nn.Linear running fp8 GEMMs when available (bias in bf16)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16
            and x.shape[-1] % 16 == 0
            and self.out_features % 16 == 0
            and fp8_available()
        ):
            shape = x.shape[:-1] + (self.out_features,)
            flat = x.reshape(-1, x.shape[-1])
            if flat.shape[0] % 16 == 0:
                y = _FP8MMFn.apply(flat, self.weight).reshape(shape)
                if self.bias is not None:
                    y = y + self.bias
                return y
        return super().forward(x)


def convert_linears_to_fp8(model: nn.Module, min_features: int = 1024,
                           delayed: bool = True) -> int:
    """Swap every big-enough nn.Linear for an fp8 linear in place.

    ``delayed=True`` (default) uses :class:`FP8LinearDelayed` — scales come
    from the previous step's amax (no extra reduction pass; the next amax is
    a byproduct of the fused cast kernel) and the backward's column-major
    operands come from the dual-layout quantizer.  ``delayed=False`` keeps
    the per-call-amax v1 path.  Returns the number of swapped layers.
    """
    if delayed:
        from stoke.nn.fp8_delayed import FP8LinearDelayed as cls
    else:
        cls = FP8Linear
    n = 0
    for mod in model.modules():
        for name, child in list(mod.named_children()):
            if (
                type(child) is nn.Linear
                and child.in_features >= min_features
                and child.in_features % 16 == 0
                and child.out_features % 16 == 0
            ):
                fp8 = cls(child.in_features, child.out_features,
                          bias=child.bias is not None)
                fp8.weight = child.weight
                if child.bias is not None:
                    fp8.bias = child.bias
                setattr(mod, name, fp8)
                n += 1
    return n
