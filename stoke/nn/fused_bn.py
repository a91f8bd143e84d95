# -*- coding: utf-8 -*-
"""Fused BatchNorm(+residual)(+ReLU) module backed by the gfx950 HIP kernels.

Profiling (profiles/resnet50_b256_steady_r01.txt) showed MIOpen BatchNorm +
eager residual-add/ReLU consume ~50% of a ResNet-50 bf16 step on MI355X.
``FusedBNAct2d`` collapses the whole bn -> (+residual) -> relu chain into two
streaming kernels forward (stats, apply) and two backward (reduce with the
ReLU mask folded in, apply) — see ``csrc/fused_bn.hip``.

State-dict compatible with ``torch.nn.BatchNorm2d`` (weight, bias,
running_mean, running_var, num_batches_tracked).  Falls back to eager
``F.batch_norm`` on CPU / non-bf16 / non-channels-last inputs, so CPU tests
and fp32 runs behave identically to the textbook composition.
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _can_fuse(x: torch.Tensor) -> bool:
    from stoke import ops

    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and x.dim() == 4
        and x.shape[1] % 8 == 0
        and x.shape[1] <= 4096
        and (256 % (x.shape[1] // 8) == 0 or x.shape[1] // 8 == 256)
        and x.is_contiguous(memory_format=torch.channels_last)
        and ops.has_ext()
    )


def _as_2d(x: torch.Tensor) -> torch.Tensor:
    # channels_last NCHW storage IS [N*H*W, C] row-major
    n, c, h, w = x.shape
    return x.permute(0, 2, 3, 1).reshape(n * h * w, c)


def _as_4d(flat: torch.Tensor, like: torch.Tensor) -> torch.Tensor:
    n, c, h, w = like.shape
    return flat.view(n, h, w, c).permute(0, 3, 1, 2)


class _FusedBNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                eps, momentum, relu):
        from stoke import _C

        x2d = _as_2d(x)
        r2d = _as_2d(residual) if residual is not None else None
        y2d, mean, invstd = _C.bn_fwd_train(
            x2d, r2d, weight, bias, running_mean, running_var, eps, momentum,
            relu,
        )
        ctx.save_for_backward(x2d, y2d, mean, invstd, weight)
        ctx.relu = relu
        ctx.has_res = residual is not None
        ctx.like_shape = x.shape
        return _as_4d(y2d, x)

    @staticmethod
    def backward(ctx, dy):
        from stoke import _C

        x2d, y2d, mean, invstd, weight = ctx.saved_tensors
        if not dy.is_contiguous(memory_format=torch.channels_last):
            dy = dy.contiguous(memory_format=torch.channels_last)
        dy2d = _as_2d(dy)
        dx2d, dgamma, dbeta, dres2d = _C.bn_bwd(
            x2d, dy2d, y2d, mean, invstd, weight, ctx.relu, ctx.has_res
        )
        dx = _as_4d(dx2d, dy)
        dres = _as_4d(dres2d, dy) if ctx.has_res else None
        return dx, dres, dgamma, dbeta, None, None, None, None, None


class FusedBNAct2d(nn.Module):
    """BatchNorm2d fused with optional residual add and ReLU."""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1, relu: bool = True):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer(
            "num_batches_tracked", torch.tensor(0, dtype=torch.long)
        )

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        if _can_fuse(x) and (residual is None or _can_fuse(residual)):
            if self.training:
                self.num_batches_tracked += 1
                return _FusedBNFn.apply(
                    x, residual, self.weight, self.bias, self.running_mean,
                    self.running_var, self.eps, self.momentum, self.relu,
                )
            from stoke import _C

            y2d = _C.bn_fwd_eval(
                _as_2d(x),
                _as_2d(residual) if residual is not None else None,
                self.weight, self.bias, self.running_mean, self.running_var,
                self.eps, self.relu,
            )
            return _as_4d(y2d, x)
        # Eager fallback (CPU / fp32 / odd layouts) — identical math
        out = F.batch_norm(
            x, self.running_mean, self.running_var, self.weight, self.bias,
            self.training, self.momentum, self.eps,
        )
        if residual is not None:
            out = out + residual
        return F.relu(out, inplace=True) if self.relu else out

    def extra_repr(self):
        return (f"{self.num_features}, eps={self.eps}, "
                f"momentum={self.momentum}, relu={self.relu}")
