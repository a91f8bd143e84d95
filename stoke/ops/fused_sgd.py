# -*- coding: utf-8 -*-
"""Fused SGD-momentum optimizer backed by the native HIP multi-tensor kernel.

The MI355X-native replacement for apex/fairscale FusedSGD, which the
reference detects by class name for its zero_grad semantics
(``stoke/utils.py:103``; SURVEY.md section 2.3 names fused SGD next to
fused Adam in the native work-list).  One kernel launch updates every
parameter per dtype group; scaler integration keeps the skip-on-overflow
decision on device.
"""

from typing import Optional

import torch

from stoke import ops


class FusedSGD(torch.optim.Optimizer):
    """torch.optim.SGD semantics (momentum/dampening/nesterov/weight decay),
    multi-tensor fused on ROCm.  bf16 params get fp32 master weights and
    fp32 momentum buffers (pure-bf16-weights LM path)."""

    step_supports_found_inf = True
    zero_grad_prefers_none = True

    def __init__(self, params, lr: float = 1e-3, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError(
                "Nesterov momentum requires a momentum and zero dampening")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None,
             found_inf: Optional[torch.Tensor] = None,
             inv_scale: Optional[torch.Tensor] = None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            momentum = group["momentum"]
            # One launch per (dtype, first-step) batch: the kernel's
            # first_step flag implements torch's buf-init (buf = d_p), and
            # a param whose state appears later (e.g. unfrozen mid-run)
            # must not drag the whole group back to first-step semantics.
            batches = {}  # (is_bf16, first) -> [p, g, b, w]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                first = len(state) == 0
                if first:
                    state["momentum_buffer"] = (
                        torch.zeros_like(p, dtype=torch.float32)
                        if momentum != 0.0
                        else torch.zeros(0, dtype=torch.float32,
                                         device=p.device)
                    )
                    if p.dtype == torch.bfloat16:
                        state["master"] = p.detach().float()
                is_bf16 = p.dtype == torch.bfloat16
                b = batches.setdefault((is_bf16, first), ([], [], [], []))
                b[0].append(p.data)
                b[1].append(p.grad.data)
                b[2].append(state["momentum_buffer"])
                if is_bf16:
                    b[3].append(state["master"])
            for (is_bf16, first), (ps, gs, bs, ws) in batches.items():
                # momentum==0 passes params again in the unused depth-3
                # slot so the multi-tensor plumbing stays uniform
                ops.fused_sgd_(
                    ps, gs, bs if momentum != 0.0 else ps,
                    group["lr"], momentum, group["dampening"],
                    group["weight_decay"], group["nesterov"], first,
                    found_inf=found_inf, inv_scale=inv_scale,
                    masters=ws if is_bf16 else None,
                )
        return loss
