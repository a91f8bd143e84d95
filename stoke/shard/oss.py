# -*- coding: utf-8 -*-
"""Optimizer-state sharding (ZeRO-1) over RCCL.

In-house replacement for fairscale ``OSS`` (reference wrap at
``stoke/extensions.py:109-141``): parameters are partitioned across ranks by
greedy size balancing; each rank runs the inner optimizer (typically the HIP
``FusedAdamW``) only on its shard, then the updated shards are broadcast in
flat ~128 MB buckets — on the full xGMI mesh every rank's broadcast rides a
different link pair, so the phase approaches all-gather bandwidth.

Optimizer state memory per rank drops by ~world_size; gradients stay
replicated (pair with the SDDP engine for grad sharding = ZeRO-2).
"""

from typing import Dict, List, Optional, Type

import torch
import torch.distributed as dist

from stoke.comm import StokeProcessGroup


class OSSOptimizer(torch.optim.Optimizer):
    """Rank-sharded wrapper around an inner optimizer class.

    ``param_groups`` exposes the FULL parameter set (so grad clipping and
    scaler unscale see every gradient, which are replicated after DDP
    all-reduce); the inner optimizer only holds this rank's partition.
    """

    def __init__(
        self,
        params,
        optim: Type[torch.optim.Optimizer],
        pg: StokeProcessGroup,
        broadcast_fp16: bool = False,
        bucket_bytes: int = 128 * 1024 * 1024,
        **optim_kwargs,
    ):
        self._pg = pg
        self._broadcast_fp16 = broadcast_fp16
        self._bucket_bytes = bucket_bytes
        # Normalize to param-group dicts
        param_groups = list(params)
        if len(param_groups) == 0:
            raise ValueError("OSSOptimizer got an empty parameter list")
        if not isinstance(param_groups[0], dict):
            param_groups = [{"params": param_groups}]
        super().__init__(param_groups, optim_kwargs)
        # Greedy balanced partition of every group's params across ranks
        self._rank_params: List[List[torch.nn.Parameter]] = [
            [] for _ in range(pg.world_size)
        ]
        self._owner: Dict[int, int] = {}
        sizes = [0] * pg.world_size
        inner_groups = [
            {k: v for k, v in g.items() if k != "params"} for g in self.param_groups
        ]
        for gi, group in enumerate(self.param_groups):
            inner_groups[gi]["params"] = []
            for p in group["params"]:
                r = sizes.index(min(sizes))
                sizes[r] += p.numel()
                self._owner[id(p)] = r
                self._rank_params[r].append(p)
                if r == pg.rank:
                    inner_groups[gi]["params"].append(p)
        self.optim = optim(
            [g for g in inner_groups if len(g["params"]) > 0] or
            [{**inner_groups[0], "params": []}],
            **optim_kwargs,
        )
        # Mirror supports-found-inf from the inner optimizer (HIP FusedAdamW)
        self.step_supports_found_inf = getattr(
            self.optim, "step_supports_found_inf", False
        )

    # ---------------------------------------------------------------- owner
    def param_owner(self, p: torch.nn.Parameter) -> int:
        return self._owner[id(p)]

    @property
    def rank_params(self) -> List[List[torch.nn.Parameter]]:
        return self._rank_params

    # ----------------------------------------------------------------- step
    @torch.no_grad()
    def step(self, closure=None, **kwargs):
        loss = self.optim.step(closure=closure, **kwargs) if closure else \
            self.optim.step(**kwargs)
        if self._pg.world_size > 1:
            self._broadcast_shards()
        return loss

    def _broadcast_shards(self):
        """Broadcast each rank's updated parameter shard in flat buckets."""
        for r in range(self._pg.world_size):
            bucket: List[torch.nn.Parameter] = []
            nbytes = 0
            for p in self._rank_params[r]:
                bucket.append(p)
                nbytes += p.numel() * p.element_size()
                if nbytes >= self._bucket_bytes:
                    self._broadcast_bucket(bucket, r)
                    bucket, nbytes = [], 0
            if bucket:
                self._broadcast_bucket(bucket, r)

    def _broadcast_bucket(self, bucket: List[torch.nn.Parameter], src: int):
        by_dtype: Dict[torch.dtype, List[torch.nn.Parameter]] = {}
        for p in bucket:
            by_dtype.setdefault(p.dtype, []).append(p)
        for dt, ps in by_dtype.items():
            comm_dt = (
                torch.float16
                if self._broadcast_fp16 and dt == torch.float32
                else dt
            )
            flat = torch.cat([p.data.reshape(-1).to(comm_dt) for p in ps])
            dist.broadcast(flat, src=src)
            if self._pg.rank != src:
                offset = 0
                for p in ps:
                    n = p.numel()
                    p.data.copy_(flat[offset : offset + n].view_as(p).to(dt))
                    offset += n

    # ----------------------------------------------------------- grad utils
    def zero_grad(self, set_to_none: bool = True):
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    if set_to_none:
                        p.grad = None
                    else:
                        p.grad.zero_()

    def clip_grad_norm(
        self, max_norm: float, norm_type: float = 2.0, grads_sharded: bool = False
    ) -> torch.Tensor:
        """Global-norm clip.

        With replicated grads (DDP+OSS) the norm is computed locally over all
        grads (identical on every rank).  With owner-sharded grads (SDDP) the
        per-rank partial norm^p is all-reduced first (reference behavior of
        fairscale ``OSS.clip_grad_norm``, called at ``fp16.py:228``).
        """
        from stoke import ops

        if grads_sharded:
            grads = [
                p.grad
                for p in self._rank_params[self._pg.rank]
                if p.grad is not None
            ]
        else:
            grads = [
                p.grad
                for g in self.param_groups
                for p in g["params"]
                if p.grad is not None
            ]
        if norm_type == 2.0:
            if grads:
                total_sq = ops.multi_tensor_l2norm(grads).pow(2)
            else:
                total_sq = torch.zeros(1, device=self._pg.device)
            if grads_sharded and self._pg.world_size > 1:
                dist.all_reduce(total_sq)
            total_norm = total_sq.sqrt()
        else:
            local = (
                torch.stack([g.norm(norm_type) for g in grads]).pow(norm_type).sum()
                if grads
                else torch.zeros((), device=self._pg.device)
            )
            if grads_sharded and self._pg.world_size > 1:
                dist.all_reduce(local)
            total_norm = local.pow(1.0 / norm_type).reshape(1)
        clip_coef = max_norm / (total_norm + 1e-6)
        coef = torch.clamp(clip_coef, max=1.0)
        all_grads = [
            p.grad for g in self.param_groups for p in g["params"] if p.grad is not None
        ]
        ops.multi_tensor_scale_(all_grads, coef.float())
        return total_norm

    # ---------------------------------------------------------- checkpointing
    def state_dict(self) -> dict:
        """Local shard state only; use consolidate_state_dict for full state."""
        return {"inner": self.optim.state_dict(), "rank": self._pg.rank}

    def consolidate_state_dict(self, recipient_rank: int = 0) -> Optional[dict]:
        """Gather the full (world-size-independent) optimizer state.

        Keys of the returned dict follow torch optimizer state_dict layout
        with param indices over the FULL parameter list, so a checkpoint can
        be reloaded at any world size (reference contract: SURVEY.md 5.4).
        """
        all_params = [p for g in self.param_groups for p in g["params"]]
        index_of = {id(p): i for i, p in enumerate(all_params)}
        # Map local shard state to global indices with CPU tensors
        local_inner = self.optim.state_dict()
        local_params = [
            p for g in self.optim.param_groups for p in g["params"]
        ]
        local_state = {}
        for li, st in local_inner.get("state", {}).items():
            gp = local_params[li]
            local_state[index_of[id(gp)]] = {
                k: (v.cpu() if isinstance(v, torch.Tensor) else v)
                for k, v in st.items()
            }
        gathered = self._pg.all_gather_object(local_state)
        if self._pg.rank != recipient_rank and recipient_rank >= 0:
            return None
        full_state = {}
        for shard in gathered:
            full_state.update(shard)
        # Emit FULL hyperparameters per group (inner-optimizer defaults +
        # per-group overrides): the outer param_groups only carry the kwargs
        # the user passed, and a checkpoint that omitted e.g. Adam's betas
        # could not be loaded into a plain torch optimizer at another world
        # size (the world-size-independence contract).
        return {
            "state": full_state,
            "param_groups": [
                {
                    **self.optim.defaults,
                    **{k: v for k, v in g.items() if k != "params"},
                    "params": [index_of[id(p)] for p in g["params"]],
                }
                for g in self.param_groups
            ],
        }

    def load_full_state_dict(self, full: dict):
        """Load this rank's slice out of a consolidated state dict."""
        all_params = [p for g in self.param_groups for p in g["params"]]
        local_params = [p for g in self.optim.param_groups for p in g["params"]]
        local_index = {id(p): i for i, p in enumerate(local_params)}
        state = {}
        for gi, st in full.get("state", {}).items():
            p = all_params[int(gi)]
            if id(p) in local_index:
                state[local_index[id(p)]] = {
                    k: (v.to(p.device) if isinstance(v, torch.Tensor) else v)
                    for k, v in st.items()
                }
        inner_sd = {
            "state": state,
            "param_groups": [
                {
                    **{k: v for k, v in g.items() if k != "params"},
                    "params": list(range(len(g["params"]))),
                }
                for g in self.optim.param_groups
            ],
        }
        # Re-index param_groups' params per inner group layout
        offset = 0
        pgs = []
        for g in self.optim.param_groups:
            n = len(g["params"])
            pgs.append(
                {
                    **{k: v for k, v in g.items() if k != "params"},
                    "params": list(range(offset, offset + n)),
                }
            )
            offset += n
        inner_sd["param_groups"] = pgs
        self.optim.load_state_dict(inner_sd)
