# -*- coding: utf-8 -*-
"""Fused AdamW optimizer backed by the native HIP multi-tensor kernel.

The MI355X-native replacement for the apex/deepspeed/fairscale FusedAdam the
reference detects by name (``stoke/utils.py:103``):
one kernel launch updates every parameter (per dtype group), with optional
scaler integration (the kernel itself skips the step when found_inf is set,
so no host sync is needed on the skip path).
"""

import math
from typing import Optional

import torch

from stoke import ops


class FusedAdamW(torch.optim.Optimizer):
    """AdamW with decoupled weight decay; multi-tensor fused on ROCm.

    The class name contains "Fused" on purpose: zero_grad keeps gradients
    allocated (reference ``utils.py:103-106`` semantics).
    """

    # StokeGradScaler passes found_inf to step() so the skip-on-overflow
    # decision stays on device (no host sync).
    step_supports_found_inf = True
    # Multi-tensor-apply re-collects grads each step, so set_to_none zeroing
    # is safe and skips the fillBuffer pass classic fused optimizers need
    # (1.7% of a ResNet-50 step in profiles/resnet50_b256_steady_r01.txt).
    zero_grad_prefers_none = True

    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 1e-2,
        adam_w_mode: bool = True,
        offload_state: bool = False,
        offload_path: Optional[str] = None,
    ):
        """``offload_state=True`` keeps exp_avg/exp_avg_sq (and the fp32
        master for bf16 params) in pinned host memory — the in-house
        equivalent of DeepSpeed's CPU optimizer offload
        (``DeepspeedOffloadOptimizerConfig``, reference ``configs.py:308-342``).
        Each step streams state H2D, runs the fused HIP kernel, and streams
        it back D2H (pinned + non_blocking, ordered on the current stream);
        HBM then only holds params+grads, trading step time for capacity.

        ``offload_path`` additionally backs the state with FILES under that
        directory (``torch.from_file`` shared mappings) — the NVMe offload
        tier (``device="nvme"`` + ``nvme_path``, reference
        ``configs.py:308-342``): state pages live in the page cache and
        spill to disk under memory pressure instead of pinning host RAM."""
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            adam_w_mode=adam_w_mode,
        )
        self.offload_state = offload_state
        self.offload_path = offload_path
        if offload_path is not None:
            import os

            os.makedirs(offload_path, exist_ok=True)
            self._offload_seq = 0
        super().__init__(params, defaults)

    def _host_state_tensor(self, p) -> torch.Tensor:
        """fp32 host tensor for one state slot: pinned RAM, or a shared
        file mapping under offload_path (NVMe tier)."""
        if self.offload_path is None:
            return torch.zeros(
                p.shape, dtype=torch.float32, device="cpu",
                pin_memory=p.is_cuda,
            )
        import os

        fn = os.path.join(self.offload_path,
                          f"adamw_state_{self._offload_seq}.bin")
        self._offload_seq += 1
        t = torch.from_file(fn, shared=True, size=p.numel(),
                            dtype=torch.float32).view(p.shape)
        t.zero_()
        return t

    @torch.no_grad()
    def step(
        self,
        closure=None,
        found_inf: Optional[torch.Tensor] = None,
        inv_scale: Optional[torch.Tensor] = None,
    ):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        if self.offload_state:
            return self._step_offload(loss, found_inf, inv_scale)
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            wd = group["weight_decay"] if group["adam_w_mode"] else 0.0
            fp32_p, fp32_g, fp32_m, fp32_v = [], [], [], []
            bf16_p, bf16_g, bf16_m, bf16_v, bf16_w = [], [], [], [], []
            step_t = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype == torch.bfloat16:
                        state["master"] = p.detach().float()
                state["step"] += 1
                step_t = state["step"]
                if p.dtype == torch.bfloat16:
                    bf16_p.append(p.data)
                    bf16_g.append(p.grad.data)
                    bf16_m.append(state["exp_avg"])
                    bf16_v.append(state["exp_avg_sq"])
                    bf16_w.append(state["master"])
                else:
                    fp32_p.append(p.data)
                    fp32_g.append(p.grad.data)
                    fp32_m.append(state["exp_avg"])
                    fp32_v.append(state["exp_avg_sq"])
            if step_t is None:
                continue
            if fp32_p:
                ops.fused_adamw_(
                    fp32_p, fp32_g, fp32_m, fp32_v, step_t,
                    group["lr"], beta1, beta2, group["eps"], wd,
                    found_inf=found_inf, inv_scale=inv_scale,
                )
            if bf16_p:
                ops.fused_adamw_(
                    bf16_p, bf16_g, bf16_m, bf16_v, step_t,
                    group["lr"], beta1, beta2, group["eps"], wd,
                    found_inf=found_inf, inv_scale=inv_scale, masters=bf16_w,
                )
        return loss

    def _step_offload(self, loss, found_inf, inv_scale):
        """Pinned-host state variant: per-parameter H2D -> fused kernel -> D2H.

        All copies are non_blocking on the current stream, so kernel K of
        parameter i overlaps the H2D of parameter i+1 at the copy-engine
        level while strict stream ordering keeps the math correct.
        """
        on_gpu = any(
            p.is_cuda for g in self.param_groups for p in g["params"]
        )
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            wd = group["weight_decay"] if group["adam_w_mode"] else 0.0
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = self._host_state_tensor(p)
                    state["exp_avg_sq"] = self._host_state_tensor(p)
                    if p.dtype == torch.bfloat16:
                        if self.offload_path is not None:
                            w = self._host_state_tensor(p)
                            w.copy_(p.detach().float().cpu())
                            state["master"] = w
                        else:
                            m = p.detach().float().cpu()
                            state["master"] = m.pin_memory() if on_gpu else m
                state["step"] += 1
                m_dev = state["exp_avg"].to(p.device, non_blocking=True)
                v_dev = state["exp_avg_sq"].to(p.device, non_blocking=True)
                if p.dtype == torch.bfloat16:
                    w_dev = state["master"].to(p.device, non_blocking=True)
                    ops.fused_adamw_(
                        [p.data], [p.grad.data], [m_dev], [v_dev],
                        state["step"], group["lr"], beta1, beta2,
                        group["eps"], wd, found_inf=found_inf,
                        inv_scale=inv_scale, masters=[w_dev],
                    )
                    state["master"].copy_(w_dev, non_blocking=True)
                else:
                    ops.fused_adamw_(
                        [p.data], [p.grad.data], [m_dev], [v_dev],
                        state["step"], group["lr"], beta1, beta2,
                        group["eps"], wd, found_inf=found_inf,
                        inv_scale=inv_scale,
                    )
                state["exp_avg"].copy_(m_dev, non_blocking=True)
                state["exp_avg_sq"].copy_(v_dev, non_blocking=True)
        # Host buffers must not be reused before their D2H copies land.
        if on_gpu:
            torch.cuda.synchronize()
        return loss
