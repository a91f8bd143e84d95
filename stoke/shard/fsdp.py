# -*- coding: utf-8 -*-
"""Fully-sharded data parallel (ZeRO-3) over RCCL — in-house flat-param engine.

Replaces fairscale ``FullyShardedDataParallel`` (reference wrap at
``stoke/extensions.py:319-376``) with an engine designed for the 8x MI355X
node: parameters live as rank-local flat fp32 shards (1/world_size each);
full parameters are materialized per unit by an RCCL all-gather immediately
before that unit's forward/backward and freed immediately after, and
gradients leave backward through a single reduce-scatter per unit — the
per-link-efficient primitives on the full xGMI mesh.

Memory model per GPU (288 GB HBM3E): fp32 shard (4N/W bytes) + fp32 optimizer
state (8N/W) + transient full bf16 params of the LARGEST unit only
(2N_unit) + bf16 activation working set — which is what makes O(10B+)
parameters per GPU trainable at world size 8.

Structure:
* auto-wrap: every submodule with >= ``min_wrap_params`` parameters becomes a
  shard unit (recursing through containers); remaining params form the root
  unit (gathered for the whole step, per ``disable_reshard_on_root``).
* mixed precision: shard master stays fp32; all-gather and compute run in
  ``compute_dtype`` (bf16 by default on CDNA4 — no loss scaler needed);
  ``fp32_reduce_scatter`` optionally reduces grads in fp32.
* the optimizer sees ONE flat shard parameter per unit (``parameters()`` on
  the wrapper yields only shards), so the HIP FusedAdamW updates each unit
  in a single multi-tensor launch.
"""

import math
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from stoke.comm import StokeProcessGroup

_ALIGN = 4  # element alignment so flat views stay 16-B aligned for float4


def _pad(n: int, mult: int) -> int:
    return ((n + mult - 1) // mult) * mult


class _ShardUnit:
    """One flat-parameter shard: a submodule's params flattened and sharded."""

    def __init__(
        self,
        name: str,
        module: nn.Module,
        params: List[nn.Parameter],
        param_names: List[str],
        pg: StokeProcessGroup,
        compute_dtype: torch.dtype,
        fp32_reduce_scatter: bool,
        reshard_after_forward: bool,
    ):
        self.name = name
        self.module = module
        self.params = params
        self.param_names = param_names
        self._pg = pg
        self.compute_dtype = compute_dtype
        self.fp32_reduce_scatter = fp32_reduce_scatter
        self.reshard_after_forward = reshard_after_forward
        self.offsets: List[int] = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += _pad(p.numel(), _ALIGN)
        self.total = off
        self.padded = _pad(max(off, 1), pg.world_size * _ALIGN)
        self.shard_nelem = self.padded // pg.world_size
        # Build the fp32 master shard from current param values
        device = params[0].device
        flat = torch.zeros(self.padded, dtype=torch.float32, device=device)
        for p, o in zip(params, self.offsets):
            flat[o : o + p.numel()].copy_(p.data.reshape(-1).float())
        if pg.world_size > 1:
            dist.broadcast(flat, src=0)  # startup consistency
        r = pg.rank
        self.shard = nn.Parameter(
            flat[r * self.shard_nelem : (r + 1) * self.shard_nelem].clone()
        )
        del flat
        # Free the originals; data will be views of the gathered buffer
        # One persistent full buffer whose storage is resized 0 <-> full:
        # autograd-saved views (conv/linear weights) reference this storage,
        # so resize-to-zero is what actually releases HBM after forward while
        # the pre-backward re-gather makes the same views valid again
        # (the standard flat-param storage trick; torch/fairscale FSDP do the
        # same — re-implemented here, not imported).
        self.full_flat = torch.empty(
            self.padded, dtype=compute_dtype, device=device
        )
        self._full_bytes = self.full_flat.untyped_storage().nbytes()
        for p, o in zip(self.params, self.offsets):
            p.data = self.full_flat[o : o + p.numel()].view(p._orig_shape)
        self.materialized = True
        self._gather_event = None
        self._gather_pending = False
        self._rs_event = None
        self._pending_shard_g = None
        self.free_full()
        self.grad_ready = 0
        self.rs_done = False

    # ------------------------------------------------------------- gather
    def gather(self, comm_stream=None):
        """Materialize full params via RCCL all-gather.

        With ``comm_stream`` the collective runs on a side stream so the
        NEXT unit's gather overlaps the CURRENT unit's compute (forward and
        backward prefetch — SURVEY.md section 7 hard-part #1).  Consumers
        must call :meth:`wait_gather` on the compute stream first.
        """
        if self.materialized:
            return
        # Storage (re)allocated on the compute stream: the allocator then
        # owns the block on the stream that also frees it (free_full), and
        # the comm stream's writes are fenced by record_stream below.
        self.full_flat.untyped_storage().resize_(self._full_bytes)
        self.materialized = True
        if comm_stream is not None:
            comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(comm_stream):
                comm_shard = self.shard.data.to(self.compute_dtype)
                if self._pg.world_size > 1:
                    self._pg.all_gather_flat(self.full_flat, comm_shard)
                else:
                    self.full_flat.copy_(comm_shard)
                self.full_flat.record_stream(comm_stream)
            if self._gather_event is None:
                self._gather_event = torch.cuda.Event()
            self._gather_event.record(comm_stream)
            self._gather_pending = True
            return
        comm_shard = self.shard.data.to(self.compute_dtype)
        if self._pg.world_size > 1:
            self._pg.all_gather_flat(self.full_flat, comm_shard)
        else:
            self.full_flat.copy_(comm_shard)

    def wait_gather(self):
        """Compute-stream fence for an async :meth:`gather`."""
        if self._gather_pending:
            torch.cuda.current_stream().wait_event(self._gather_event)
            self._gather_pending = False

    def free_full(self):
        if not self.materialized:
            return
        self.wait_gather()
        self.full_flat.untyped_storage().resize_(0)
        self.materialized = False

    # ------------------------------------------------------- grad handling
    def reduce_scatter_grads(self, average: bool = True, comm_stream=None):
        """Pack full grads -> one reduce-scatter -> fp32 shard gradient.

        Async variant (``comm_stream`` given): the pack runs on the compute
        stream, the reduce-scatter on the side stream (overlapping the rest
        of backward), and :meth:`finalize_grad` converts/accumulates after
        an event wait.
        """
        if self.rs_done:
            return
        self.rs_done = True
        device = self.shard.device
        rs_dtype = (
            torch.float32 if self.fp32_reduce_scatter else self.compute_dtype
        )
        flat_g = torch.zeros(self.padded, dtype=rs_dtype, device=device)
        for p, o in zip(self.params, self.offsets):
            if p.grad is not None:
                flat_g[o : o + p.numel()].copy_(
                    p.grad.reshape(-1).to(rs_dtype)
                )
            p.grad = None
        if comm_stream is not None:
            comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(comm_stream):
                shard_g = torch.empty(
                    self.shard_nelem, dtype=rs_dtype, device=device
                )
                if self._pg.world_size > 1:
                    self._pg.reduce_scatter_flat(shard_g, flat_g)
                else:
                    shard_g.copy_(flat_g)
                flat_g.record_stream(comm_stream)
            if self._rs_event is None:
                self._rs_event = torch.cuda.Event()
            self._rs_event.record(comm_stream)
            self._pending_shard_g = shard_g
            self.grad_ready = 0
            self.free_full()
            return
        shard_g = torch.empty(self.shard_nelem, dtype=rs_dtype, device=device)
        if self._pg.world_size > 1:
            self._pg.reduce_scatter_flat(shard_g, flat_g)
        else:
            shard_g.copy_(flat_g)
        del flat_g
        self._pending_shard_g = shard_g
        self.grad_ready = 0
        self.free_full()
        self.finalize_grad(average)

    def finalize_grad(self, average: bool = True):
        """Convert/accumulate the reduce-scattered gradient on the compute
        stream (after waiting on the side-stream event if one is pending)."""
        shard_g = self._pending_shard_g
        if shard_g is None:
            return
        self._pending_shard_g = None
        if self._rs_event is not None:
            torch.cuda.current_stream().wait_event(self._rs_event)
            # last read of the comm-stream product happens on compute below
            shard_g.record_stream(torch.cuda.current_stream())
        if average and self._pg.world_size > 1:
            shard_g = shard_g.float() / self._pg.world_size
        else:
            shard_g = shard_g.float()
        if self.shard.grad is None:
            self.shard.grad = shard_g
        else:
            self.shard.grad.add_(shard_g)


class StokeFSDPModule(nn.Module):
    """ZeRO-3 wrapper: per-unit all-gather / reduce-scatter around compute."""

    def __init__(
        self,
        module: nn.Module,
        pg: StokeProcessGroup,
        compute_dtype: Optional[torch.dtype] = None,
        mixed_precision: bool = False,
        fp32_reduce_scatter: bool = False,
        reshard_after_forward: bool = True,
        disable_reshard_on_root: bool = True,
        min_wrap_params: int = 1_000_000,
        flatten_parameters: bool = True,
    ):
        super().__init__()
        self.module = module
        self._pg = pg
        if compute_dtype is None:
            compute_dtype = torch.bfloat16 if mixed_precision else torch.float32
        self.compute_dtype = compute_dtype
        self._reshard_after_forward = reshard_after_forward
        self.require_backward_grad_sync = True
        # Record original shapes/names before flattening
        for n, p in module.named_parameters():
            p._orig_shape = p.shape
            p._orig_name = n
        if pg.world_size > 1:
            pg.broadcast_module_states(module)  # buffers stay replicated
        self.units: List[_ShardUnit] = []
        assigned: set = set()
        if flatten_parameters:
            self._auto_wrap(module, "", assigned, min_wrap_params,
                            fp32_reduce_scatter)
        # Root unit with everything unassigned
        rest = [
            (n, p)
            for n, p in module.named_parameters()
            if id(p) not in assigned and p.requires_grad
        ]
        if rest:
            unit = _ShardUnit(
                "(root)", module, [p for _, p in rest], [n for n, _ in rest],
                pg, compute_dtype, fp32_reduce_scatter,
                reshard_after_forward=not disable_reshard_on_root,
            )
            self.units.append(unit)
        self._unit_of_param: Dict[int, _ShardUnit] = {}
        for u in self.units:
            for p in u.params:
                self._unit_of_param[id(p)] = u
        self._register_hooks()
        self._callback_queued = False
        # Side stream for all-gather prefetch and async reduce-scatter
        # (RCCL only; gloo/CPU stays synchronous).  Unit execution order is
        # recorded on the first forward and drives depth-1 prefetch after.
        self._comm_stream = (
            torch.cuda.Stream()
            if pg.backend == "nccl" and torch.cuda.is_available()
            else None
        )
        self._fwd_order: List[_ShardUnit] = []
        self._order_final = False

    def _auto_wrap(self, root, prefix, assigned, min_params, fp32_rs):
        """Create a shard unit per sufficiently large submodule."""
        for cname, child in root.named_children():
            path = f"{prefix}{cname}"
            if isinstance(child, (nn.ModuleList, nn.Sequential, nn.ModuleDict)):
                self._auto_wrap(child, path + ".", assigned, min_params, fp32_rs)
                continue
            plist = [
                (n, p)
                for n, p in child.named_parameters()
                if p.requires_grad and id(p) not in assigned
            ]
            nparams = sum(p.numel() for _, p in plist)
            if nparams >= min_params and plist:
                unit = _ShardUnit(
                    path, child,
                    [p for _, p in plist],
                    [f"{path}.{n}" for n, _ in plist],
                    self._pg, self.compute_dtype, fp32_rs,
                    self._reshard_after_forward,
                )
                self.units.append(unit)
                for _, p in plist:
                    assigned.add(id(p))
            elif nparams > 0:
                # Recurse: a large grandchild may still qualify
                self._auto_wrap(child, path + ".", assigned, min_params, fp32_rs)

    # ----------------------------------------------------------------- hooks
    def _register_hooks(self):
        for u in self.units:
            if u.name != "(root)":
                u.module.register_forward_pre_hook(self._make_pre_fwd(u))
                u.module.register_forward_hook(self._make_post_fwd(u))
                u.module.register_full_backward_pre_hook(self._make_pre_bwd(u))
            for p in u.params:
                p.register_post_accumulate_grad_hook(self._grad_hook)

    def _prefetch(self, u, order):
        """Async-gather the unit AFTER ``u`` in ``order`` (depth-1 prefetch)."""
        if self._comm_stream is None or not self._order_final:
            return
        try:
            i = order.index(u)
        except ValueError:
            return
        for nxt in order[i + 1:]:
            if not nxt.materialized:
                nxt.gather(comm_stream=self._comm_stream)
                return

    def _make_pre_fwd(self, u):
        def hook(mod, args):
            if not self._order_final and u not in self._fwd_order:
                self._fwd_order.append(u)
            u.gather()  # no-op if already prefetched
            u.wait_gather()
            self._prefetch(u, self._fwd_order)
        return hook

    def _make_post_fwd(self, u):
        def hook(mod, args, out):
            if u.reshard_after_forward and self.module.training:
                u.free_full()
            elif not self.module.training and u.reshard_after_forward:
                u.free_full()
        return hook

    def _make_pre_bwd(self, u):
        def hook(mod, grad_output):
            u.gather()  # no-op if already prefetched
            u.wait_gather()
            # Backward visits units in reverse forward order
            self._prefetch(u, self._bwd_order)
        return hook

    @property
    def _bwd_order(self):
        return list(reversed(self._fwd_order))

    def _grad_hook(self, param):
        u = self._unit_of_param.get(id(param))
        if u is None:
            return
        if not self.require_backward_grad_sync:
            return  # accumulate full grads locally (no_sync)
        if not self._callback_queued:
            torch.autograd.Variable._execution_engine.queue_callback(
                self._finalize_backward
            )
            self._callback_queued = True
        u.grad_ready += 1
        if u.grad_ready == len(u.params):
            u.reduce_scatter_grads(comm_stream=self._comm_stream)

    def _finalize_backward(self):
        self._callback_queued = False
        for u in self.units:
            if not u.rs_done and any(p.grad is not None for p in u.params):
                u.reduce_scatter_grads(comm_stream=self._comm_stream)
        for u in self.units:
            u.finalize_grad()
            u.rs_done = False

    def finish_backward(self):
        """Runner-called flush: also converts no_sync-accumulated grads."""
        any_pending = any(
            (not u.rs_done) and any(p.grad is not None for p in u.params)
            for u in self.units
        )
        if self._callback_queued or any_pending:
            for u in self.units:
                if any(p.grad is not None for p in u.params):
                    u.rs_done = False
                    u.reduce_scatter_grads(comm_stream=self._comm_stream)
            self._callback_queued = False
        for u in self.units:
            u.finalize_grad()
            u.rs_done = False

    @contextmanager
    def no_sync(self):
        prev = self.require_backward_grad_sync
        self.require_backward_grad_sync = False
        try:
            yield
        finally:
            self.require_backward_grad_sync = prev

    # --------------------------------------------------------------- forward
    def forward(self, *args, **kwargs):
        for u in self.units:
            if u.name == "(root)":
                u.gather()
        # Kick off the first wrapped unit's gather on the side stream so it
        # overlaps the root gather / input embedding work.
        if self._order_final and self._comm_stream is not None:
            for u in self._fwd_order:
                if not u.materialized:
                    u.gather(comm_stream=self._comm_stream)
                    break
        out = self.module(*args, **kwargs)
        self._order_final = True
        for u in self.units:
            if u.name == "(root)" and u.reshard_after_forward and not torch.is_grad_enabled():
                u.free_full()
        return out

    # ------------------------------------------------------------ optimizer
    def parameters(self, recurse: bool = True):
        """Yield ONLY the flat shard parameters (what the optimizer updates)."""
        for u in self.units:
            yield u.shard

    def named_parameters(self, prefix: str = "", recurse: bool = True,
                         remove_duplicate: bool = True):
        for u in self.units:
            yield f"fsdp.{u.name}.shard", u.shard

    def clip_grad_norm_(self, max_norm: float, norm_type: float = 2.0):
        """Sharded global-norm clip (reference ``model.clip_grad_norm_``,
        called at ``fp16.py:231``): partial norm^p over local shards,
        all-reduced, then one scale over shard grads."""
        from stoke import ops

        grads = [u.shard.grad for u in self.units if u.shard.grad is not None]
        if norm_type == 2.0:
            total_sq = (
                ops.multi_tensor_l2norm(grads).pow(2)
                if grads
                else torch.zeros(1, device=self._pg.device)
            )
            if self._pg.world_size > 1:
                dist.all_reduce(total_sq)
            total_norm = total_sq.sqrt()
        else:
            local = (
                torch.stack([g.norm(norm_type) for g in grads]).pow(norm_type).sum()
                if grads
                else torch.zeros((), device=self._pg.device)
            )
            if self._pg.world_size > 1:
                dist.all_reduce(local)
            total_norm = local.pow(1.0 / norm_type).reshape(1)
        coef = torch.clamp(max_norm / (total_norm + 1e-6), max=1.0)
        ops.multi_tensor_scale_(grads, coef.float())
        return total_norm

    # --------------------------------------------------------- checkpointing
    def full_state_dict(self) -> dict:
        """World-size-independent model state dict (fp32), on every rank."""
        sd = {}
        for u in self.units:
            full = torch.empty(
                u.padded, dtype=torch.float32, device=u.shard.device
            )
            if self._pg.world_size > 1:
                self._pg.all_gather_flat(full, u.shard.data)
            else:
                full.copy_(u.shard.data)
            for p, name, o in zip(u.params, u.param_names, u.offsets):
                sd[name] = full[o : o + p.numel()].view(p._orig_shape).cpu().clone()
            del full
        for name, buf in self.module.named_buffers():
            sd[name] = buf.detach().cpu().clone()
        return sd

    def load_full_state_dict(self, sd: dict, strict: bool = True):
        for u in self.units:
            flat = torch.zeros(
                u.padded, dtype=torch.float32, device=u.shard.device
            )
            for p, name, o in zip(u.params, u.param_names, u.offsets):
                if name in sd:
                    flat[o : o + p.numel()].copy_(
                        sd[name].reshape(-1).float().to(u.shard.device)
                    )
                elif strict:
                    raise KeyError(f"FSDP load: missing parameter {name}")
            r = self._pg.rank
            u.shard.data.copy_(
                flat[r * u.shard_nelem : (r + 1) * u.shard_nelem]
            )
            del flat
        for name, buf in self.module.named_buffers():
            if name in sd:
                buf.data.copy_(sd[name].to(buf.device, buf.dtype))

    def gather_full_optim_state_dict(self, optimizer,
                                     rank0_only: bool = True) -> Optional[dict]:
        """Full per-original-param optimizer state.

        All ranks participate in the all-gathers; only rank 0 materializes
        the host-side dict (others return None) unless ``rank0_only=False``
        — at 8 ranks x 8B params the old every-rank materialization was a
        host-memory spike.
        """
        emit = (not rank0_only) or self._pg.rank == 0 or self._pg.world_size == 1
        full = {"state": {}, "param_groups": []}
        shard_index = {id(u.shard): i for i, u in enumerate(self.units)}
        pidx = 0
        name_index = {}
        for u in self.units:
            for name in u.param_names:
                name_index[name] = pidx
                pidx += 1
        for group in optimizer.param_groups:
            gidx = []
            for shard_p in group["params"]:
                ui = shard_index.get(id(shard_p))
                if ui is None:
                    continue
                u = self.units[ui]
                st = optimizer.state.get(shard_p, {})
                gathered = {}
                for k, v in st.items():
                    if isinstance(v, torch.Tensor) and v.numel() == u.shard_nelem:
                        fullv = torch.empty(
                            u.padded, dtype=v.dtype, device=v.device
                        )
                        if self._pg.world_size > 1:
                            self._pg.all_gather_flat(fullv, v)
                        else:
                            fullv.copy_(v)
                        gathered[k] = fullv
                    else:
                        gathered[k] = v
                for p, name, o in zip(u.params, u.param_names, u.offsets):
                    if emit:
                        entry = {}
                        for k, v in gathered.items():
                            if isinstance(v, torch.Tensor) and v.numel() == u.padded:
                                entry[k] = (
                                    v[o : o + p.numel()].view(p._orig_shape).cpu().clone()
                                )
                            else:
                                entry[k] = v
                        full["state"][name_index[name]] = entry
                    gidx.append(name_index[name])
                del gathered
            full["param_groups"].append(
                {**{k: v for k, v in group.items() if k != "params"},
                 "params": gidx}
            )
        return full if emit else None

    def load_full_optim_state_dict(self, optimizer, full: dict):
        """Re-shard a consolidated optimizer state into the local optimizer."""
        name_to_idx = {}
        for u in self.units:
            for n in u.param_names:
                name_to_idx[n] = len(name_to_idx)
        for group in optimizer.param_groups:
            for shard_p in group["params"]:
                u = next((x for x in self.units if x.shard is shard_p), None)
                if u is None:
                    continue
                # Collect per-key flats over this unit's params.  Only state
                # tensors shaped like the parameter (exp_avg, exp_avg_sq, ...)
                # re-flatten; scalar tensors (e.g. Adam's `step`) pass through.
                keys = set()
                for p, name in zip(u.params, u.param_names):
                    st = full["state"].get(name_to_idx[name])
                    if st:
                        keys.update(
                            k for k, v in st.items()
                            if isinstance(v, torch.Tensor)
                            and v.numel() == p.numel() and p.numel() > 1
                        )
                state_entry = {}
                for k in keys:
                    flat = torch.zeros(
                        u.padded, dtype=torch.float32, device=u.shard.device
                    )
                    for p, name, o in zip(u.params, u.param_names, u.offsets):
                        st = full["state"].get(name_to_idx[name])
                        if st and k in st:
                            flat[o : o + p.numel()].copy_(
                                st[k].reshape(-1).float().to(u.shard.device)
                            )
                    r = self._pg.rank
                    state_entry[k] = flat[
                        r * u.shard_nelem : (r + 1) * u.shard_nelem
                    ].clone()
                    del flat
                # Scalar entries (e.g. step) come from the first param
                st0 = full["state"].get(name_to_idx[u.param_names[0]], {})
                for k, v in st0.items():
                    if k in state_entry:
                        continue
                    if isinstance(v, torch.Tensor):
                        if v.numel() == 1:
                            state_entry[k] = v.clone()
                    else:
                        state_entry[k] = v
                optimizer.state[shard_p] = state_entry
