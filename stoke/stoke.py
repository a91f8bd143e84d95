# -*- coding: utf-8 -*-
"""The ``Stoke`` facade: the single user-facing entry point.

Public API matches the reference class (``stoke/stoke.py:49-1466``):
declarative flags + typed configs in, wrapped ``model``/``loss``/``backward``/
``step`` calls out, with automatic device placement, gradient accumulation
and clipping, loss tracking (last/agg/EMA), unified save/load and a
DataLoader shim.  Internally everything runs on the single MI355X-native
runtime (``stoke/runtime/runner.py``) instead of the reference's dynamic
mixin composition.
"""

import os
from contextlib import contextmanager, nullcontext
from typing import Callable, Dict, List, Optional, Sequence, Tuple, Type, Union
from uuid import uuid4

import torch

from torch.utils.data import Dataset
from torch.utils.data.distributed import DistributedSampler, Sampler

from stoke.configs import (
    AMPConfig,
    ApexConfig,
    ClipGradConfig,
    ClipGradNormConfig,
    DDPConfig,
    DeepspeedConfig,
    FairscaleFSDPConfig,
    FairscaleOSSConfig,
    FairscaleSDDPConfig,
    HorovodConfig,
    StokeOptimizer,
)
from stoke.data import StokeDataLoader
from stoke.ddp import StokeDDPModule
from stoke.runtime import StokeRunner
from stoke.shard import StokeFSDPModule, StokeSDDPModule
from stoke.status import DistributedOptions, FP16Options, StokeStatus
from stoke.utils import (
    ParamNormalize,
    T_co,
    _collate_fn_t,
    _worker_init_fn_t,
    zero_optimizer_grads,
)


# rocprofv3/roctracer-visible phase markers: enabled
# with STOKE_NVTX=1; torch.cuda.nvtx maps to roctx ranges on ROCm.
_NVTX = os.environ.get("STOKE_NVTX", "0") == "1"


@contextmanager
def _phase(name: str):
    if _NVTX and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(f"stoke::{name}")
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class Stoke:
    """Declarative training wrapper over the MI355X-native runtime."""

    def __init__(
        self,
        model: torch.nn.Module,
        optimizer: StokeOptimizer,
        loss: Union[Callable, List[Callable], Tuple[Callable]],
        batch_size_per_device: int,
        grad_accum_steps: Optional[int] = 1,
        grad_clip: Optional[Union[ClipGradConfig, ClipGradNormConfig]] = None,
        gpu: bool = False,
        fp16: Optional[FP16Options] = None,
        distributed: Optional[DistributedOptions] = None,
        fairscale_oss: bool = False,
        fairscale_sddp: bool = False,
        fairscale_fsdp: bool = False,
        configs: Optional[List] = None,
        info_rank: Optional[Union[int, List[int]]] = 0,
        verbose: bool = True,
        ema_weight: float = 0.1,
    ):
        self._verbose = verbose
        self._info_rank = info_rank
        self._ema_weight = ema_weight
        self._status = StokeStatus(
            batch_size_per_device=batch_size_per_device,
            grad_accum=grad_accum_steps,
            grad_clip=grad_clip,
            gpu=gpu,
            fp16=fp16,
            distributed=distributed,
            fairscale_oss=fairscale_oss,
            fairscale_sddp=fairscale_sddp,
            fairscale_fsdp=fairscale_fsdp,
            configs=configs,
        )
        self._model = self._check_model(model)
        self._optimizer = self._check_optimizer(optimizer)
        self._loss = self._check_loss(loss)
        # Build + init the unified runner
        self._runner = StokeRunner(
            status=self._status,
            loss=self._loss,
            verbose=self._verbose,
            info_rank=self._info_rank,
        )
        self._runner.setup_distributed()
        if self._verbose:
            dev_id = (
                self.rank
                if (self.rank == "cpu" or self.rank == "gpu")
                else self._info_rank
            )
            self.print(f"Printing verbose information on rank(s): {dev_id}")
        self._place_model_on_gpu()
        self._handle_ordered_wrap_ops(optimizer=optimizer)
        # Tracking vars
        self._grad_accum_counter = 0
        self._optimizer_steps = 0
        self._backward_steps = 0
        self._last_step_loss = self._set_loss_to_zero()
        self._agg_loss = self._set_loss_to_zero()
        self._rolling_mean_loss = self._set_loss_to_zero()
        self._rolling_loss_steps = 0
        self._status.set_post_init_values(world_size=self.world_size)
        # Flops profiler (in-house DeepspeedFlopsConfig equivalent): hooks
        # count forward FLOPs until profile_step optimizer steps, then print
        self._flops_profiler = None
        self._flops_cfg = None
        if (
            self._status.is_distributed_deepspeed
            and self._status.deepspeed_config.flops_profiler is not None
        ):
            from stoke.utils import FlopsProfiler

            self._flops_cfg = self._status.deepspeed_config.flops_profiler
            self._flops_profiler = FlopsProfiler(self.model_access)
            self._flops_profiler.start_profile()
        if self._verbose:
            self.print(msg=self._status)

    # ------------------------------------------------------------ wrap order
    def _wrap_optimizer_then_model(self, optimizer: StokeOptimizer):
        """Optimizer first (SDDP+OSS / apex-style / horovod): the sharded or
        master-weight optimizer must exist before the model wrapper
        (reference wrap-order rule, ``stoke.py:306-324``)."""
        self._optimizer = self._runner.build_optimizer(
            optimizer=optimizer["optimizer"],
            optimizer_kwargs=optimizer["optimizer_kwargs"],
            model=self._model,
        )
        self._runner.wrap_fp16(model=self._model, optimizer=self._optimizer)
        self._model, self._optimizer = self._runner.wrap_distributed(
            model=self._model, grad_accum=self.grad_accum, optimizer=self._optimizer
        )

    def _wrap_model_then_optimizer(self, optimizer: StokeOptimizer):
        self._model, _ = self._runner.wrap_distributed(
            model=self._model, grad_accum=self.grad_accum, optimizer=None
        )
        self._runner.wrap_fp16(model=self._model, optimizer=None)
        self._optimizer = self._runner.build_optimizer(
            optimizer=optimizer["optimizer"],
            optimizer_kwargs=optimizer["optimizer_kwargs"],
            model=self._model,
        )

    def _handle_ordered_wrap_ops(self, optimizer: StokeOptimizer):
        if (self.sharded and self.oss) or self.is_apex or self.is_horovod:
            self._wrap_optimizer_then_model(optimizer=optimizer)
        else:
            self._wrap_model_then_optimizer(optimizer=optimizer)

    # ----------------------------------------------------------- accum logic
    def _check_accum(self):
        """True on the grad-accum boundary (the backward that steps).

        The counter increments in ``backward`` BEFORE ``step`` checks, so the
        optimizer fires on the accum-th backward (reference ``stoke.py:326-334``).
        """
        return (self._grad_accum_counter + 1) % (self.grad_accum + 1) == 0

    def _check_pre_accum(self):
        return (self._grad_accum_counter + 1) % (self.grad_accum + 1) == self.grad_accum

    def _set_loss_to_zero(self):
        return (
            type(self._loss)([0.0] * len(self._loss))
            if isinstance(self._loss, (list, tuple))
            else 0.0
        )

    # -------------------------------------------------------------- EMA/print
    def reset_ema(self):
        self._rolling_mean_loss = self._set_loss_to_zero()
        self._rolling_loss_steps = 0

    def print_ema_loss(
        self, prepend_msg: str = "Current EMA Loss", single_line: bool = False
    ):
        if isinstance(self._rolling_mean_loss, (list, tuple)):
            print_vals = [
                f"{prepend_msg} {idx}: {val:.3f}"
                for idx, val in enumerate(self._rolling_mean_loss)
            ]
            self.print(print_vals, single_line=single_line)
        else:
            self.print(f"{prepend_msg}: {self._rolling_mean_loss:.3f}")

    def print_mean_accumulated_synced_loss(
        self,
        prepend_msg: str = "Mean Accumulated & Synced Loss",
        pre_backwards: bool = True,
        single_line: bool = False,
    ):
        check_fn = self._check_pre_accum if pre_backwards else self._check_accum
        if check_fn():
            if isinstance(self._agg_loss, (list, tuple)):
                print_vals = self._scale_agg_loss()
                self.print(print_vals, single_line=single_line)
            else:
                self.print(f"{prepend_msg}: {self._scale_agg_loss():.3f}")

    def _scale_agg_loss(self):
        if isinstance(self._agg_loss, (list, tuple)):
            return [val / self.grad_accum for val in self._agg_loss]
        return self._agg_loss / self.grad_accum

    def print_synced_loss(
        self,
        loss,
        prepend_msg: str = "Step Synced Loss",
        device=None,
        single_line: bool = False,
    ):
        printable_loss = self.detach_and_sync_loss(loss, device)
        if isinstance(printable_loss, (list, tuple)):
            print_vals = [
                f"{prepend_msg} {idx}: {val * self.grad_accum:.3f}"
                for idx, val in enumerate(printable_loss)
            ]
            self.print(print_vals, single_line=single_line)
        else:
            self.print(msg=f"{prepend_msg}: {printable_loss * self.grad_accum:.3f}")

    def print_on_devices(self, msg, rank: Optional[Union[int, List[int]]] = 0):
        self._runner.print_device(msg=msg, rank=rank)

    def print(self, msg, single_line: bool = False):
        self._runner.print_device(
            msg=msg, rank=self._info_rank, single_line=single_line
        )

    # ----------------------------------------------------------------- checks
    @staticmethod
    def _check_model(model: torch.nn.Module):
        if not isinstance(model, torch.nn.Module):
            raise TypeError(
                f"Stoke -- Model is not of type torch.nn.Module, currently {type(model)}"
            )
        return model

    @staticmethod
    def _check_optimizer(optimizer: StokeOptimizer):
        if not isinstance(optimizer, dict):
            raise TypeError(
                f"Stoke -- Optimizer is not of type torch.optim.Optimizer, currently {type(optimizer)}"
            )
        return optimizer

    def _check_loss(self, loss):
        if isinstance(loss, (list, tuple)):
            return type(loss)(self._check_loss(val) for val in loss)
        elif isinstance(loss, Callable):
            return loss
        raise TypeError(
            f"Stoke -- Loss is not of type Callable, currently {type(loss)}"
        )

    def _place_model_on_gpu(self):
        if self.gpu:
            if self._verbose:
                self.print("Automatically handling moving model to GPU(s)...")
            self._model.cuda()

    # ------------------------------------------------------------ DataLoader
    def DataLoader(
        self,
        dataset: Dataset[T_co],
        shuffle: bool = False,
        sampler: Optional[Sampler[int]] = None,
        batch_sampler: Optional[Sampler[Sequence[int]]] = None,
        num_workers: int = 0,
        collate_fn: _collate_fn_t = None,
        pin_memory: bool = False,
        drop_last: bool = False,
        timeout: float = 0,
        worker_init_fn: Optional[_worker_init_fn_t] = None,
        multiprocessing_context=None,
        generator=None,
        *,
        prefetch_factor: Optional[int] = None,
        persistent_workers: bool = False,
    ):
        """Build a ``StokeDataLoader`` with device placement pre-configured."""
        from stoke.data import BucketedDistributedSampler

        if self.distributed is not None and not isinstance(
            sampler, (DistributedSampler, BucketedDistributedSampler)
        ):
            raise TypeError(
                "Stoke -- Using a distributed backend requires passing an instance of a "
                "DistributedSampler to the sampler argument"
            )
        if self._verbose and self.gpu:
            self.print(
                "Stoke -- Automatically handling moving model input data to GPU(s)..."
            )
        kwargs = {
            "batch_size": self.batch_size,
            "shuffle": shuffle,
            "sampler": sampler,
            "batch_sampler": batch_sampler,
            "num_workers": num_workers,
            "collate_fn": collate_fn,
            "pin_memory": pin_memory,
            "drop_last": drop_last,
            "timeout": timeout,
            "worker_init_fn": worker_init_fn,
            "multiprocessing_context": multiprocessing_context,
            "generator": generator,
            "persistent_workers": persistent_workers,
        }
        if num_workers > 0:
            kwargs["prefetch_factor"] = (
                prefetch_factor if prefetch_factor is not None else 2
            )
        return StokeDataLoader(dataset, gpu=self.gpu, fp16=self.fp16, **kwargs)

    # --------------------------------------------------------------- hot loop
    def model(self, *args, **kwargs):
        """Forward call under the precision context."""
        with _phase("model"), self._runner.model_context:
            return self._model(*args, **kwargs)

    def loss(self, *args, **kwargs):
        """Loss call: computes, syncs for tracking, scales for accumulation."""
        with _phase("loss"), self._runner.loss_context:
            if isinstance(self._loss, (list, tuple)):
                loss = type(self._loss)(val(*args, **kwargs) for val in self._loss)
                sync_loss = [self.detach_and_sync_loss(val) for val in loss]
                self._last_step_loss = type(self._loss)(v for v in sync_loss)
                self._agg_loss = type(self._loss)(
                    self._agg_loss[idx] + val for idx, val in enumerate(sync_loss)
                )
                self._handle_ema_loss(loss=sync_loss)
                if self.grad_accum > 1 and self.model_access.training:
                    loss = type(loss)(val / self.grad_accum for val in loss)
            else:
                loss = self._loss(*args, **kwargs)
                sync_loss = self.detach_and_sync_loss(loss)
                self._last_step_loss = sync_loss
                self._agg_loss += sync_loss
                self._handle_ema_loss(loss=sync_loss)
                if self.grad_accum > 1 and self.model_access.training:
                    loss = loss / self.grad_accum
            return loss

    def _handle_ema_loss(self, loss):
        self._rolling_loss_steps += 1
        if isinstance(loss, (list, tuple)):
            self._rolling_mean_loss = type(self._rolling_mean_loss)(
                self._ema_loss(value=val, current_mean=self._rolling_mean_loss[idx])
                for idx, val in enumerate(loss)
            )
        else:
            self._rolling_mean_loss = self._ema_loss(
                value=loss, current_mean=self._rolling_mean_loss
            )

    def _ema_loss(self, value: float, current_mean: float):
        if self._rolling_loss_steps == 1:
            return value
        return (self._ema_weight * value) + ((1.0 - self._ema_weight) * current_mean)

    def backward(self, loss):
        self._grad_accum_counter += 1
        dist_cm = (
            nullcontext()
            if self._check_accum()
            else self._runner.grad_accum_context(self._model)
        )
        with _phase("backward"), dist_cm:
            self._runner.backward_call(
                loss=loss, model=self.model_access, optimizer=self._optimizer
            )
        self._backward_steps += 1

    def step(self):
        with _phase("step"):
            self._step_impl()

    def _step_impl(self):
        if self._check_accum():
            if self._verbose and self.grad_accum > 0:
                self.print(f"Gradient Accumulation Steps: {self.grad_accum}")
            if self.grad_clip is not None:
                self._runner.clip_grad(
                    self.grad_clip,
                    self._model if self.fully_sharded else self.model_access,
                    self._optimizer,
                    oss=self.oss,
                    horovod=self.is_horovod,
                    deepspeed=self.is_deepspeed,
                    fsdp=self.fully_sharded,
                )
            step_cm = (
                self._runner.step_context(self._optimizer)
                if self.grad_clip is not None
                else nullcontext()
            )
            with step_cm:
                self._runner.step_call(
                    model=self._model, optimizer=self._optimizer
                )
            self._reset()
            self._optimizer_steps += 1
            if (
                self._flops_profiler is not None
                and self._optimizer_steps == self._flops_cfg.profile_step - 1
            ):
                # zero counters so the profiled step counts exactly one
                # optimizer step's forwards
                self._flops_profiler.reset_flops()
            if (
                self._flops_profiler is not None
                and self._optimizer_steps >= self._flops_cfg.profile_step
            ):
                # rank-0 only: every rank printing (and racing on
                # output_file) was an ADVICE.md round-1 finding
                rank = self._runner.rank
                if rank in ("cpu", "gpu", 0):
                    self._flops_profiler.print_model_profile(
                        top_modules=self._flops_cfg.top_modules,
                        detailed=self._flops_cfg.detailed,
                        output_file=self._flops_cfg.output_file,
                    )
                self._flops_profiler.stop_profile()
                self._flops_profiler = None
        elif self.is_deepspeed:
            step_cm = (
                self._runner.step_context(self._optimizer)
                if self.grad_clip is not None
                else nullcontext()
            )
            with step_cm:
                self._runner.step_call(
                    model=self._model, optimizer=self._optimizer
                )

    def _reset(self):
        if self._verbose:
            self.print("Resetting all grad/variables for next optimizer step")
        if not self.is_deepspeed:
            self.zero_grads()
        self._grad_accum_counter = 0
        self._agg_loss = self._set_loss_to_zero()

    # ---------------------------------------------------------------- save/IO
    def save(
        self,
        path: str,
        name: str = None,
        extension: str = "pt",
        create_directory: bool = True,
        extras: Optional[dict] = None,
    ):
        name = name if name is not None else str(uuid4())
        out_path, tag = self._runner.save(
            model=self._model if self.fully_sharded else self.model_access,
            optimizer=self.optimizer,
            path=path,
            backward_step=self._backward_steps,
            grad_accum_step=self._grad_accum_counter,
            optimizer_step=self._optimizer_steps,
            name=name,
            scaler_dict=self.fp16_state_dict,
            extension=extension,
            create_directory=create_directory,
            extras=extras,
            status=self.status.status,
        )
        self.print(f"Successfully saved model checkpoint to {out_path}/{tag}")
        return out_path, tag

    def load(self, path: str, tag: str, strict: bool = True):
        backward_step, grad_accum_step, optimizer_step, extras = self._runner.load(
            model=self._model if self.fully_sharded else self.model_access,
            optimizer=self.optimizer,
            gpu=self.gpu,
            path=path,
            tag=tag,
            scaler_dict_fn=self._load_fp16_state_dict_fn(),
            strict=strict,
        )
        self._backward_steps = backward_step
        self._grad_accum_counter = grad_accum_step
        self._optimizer_steps = optimizer_step
        self.print(f"Successfully loaded model checkpoint from {path}/{tag}")
        return extras

    def print_num_model_parameters(
        self, normalize: ParamNormalize = ParamNormalize.MILLION
    ):
        self.print(
            f"Total Trainable Model Parameters: "
            f"{(self.num_model_parameters / normalize.value):.3f} {normalize.name}"
        )

    def detach_and_sync_loss(self, loss, device=None):
        return self._runner.detach_and_sync_loss(loss=loss, device=device)

    def zero_grads(self):
        zero_optimizer_grads(
            optimizer=self._optimizer, apex=self.is_apex, horovod=self.is_horovod
        )

    def reset(self):
        self._reset()

    def reset_tracking(self):
        self._grad_accum_counter = 0
        self._optimizer_steps = 0
        self._backward_steps = 0
        self._last_step_loss = self._set_loss_to_zero()
        self._agg_loss = self._set_loss_to_zero()
        self._rolling_mean_loss = self._set_loss_to_zero()
        self._rolling_loss_steps = 0

    def dump_model_parameter_info(self):
        self.print("Dumping all model parameter information to stdout....")
        for name, param in self.model_access.named_parameters():
            if param.requires_grad:
                self.print(
                    f"Name: {name}, Shape: {param.shape}, "
                    f"Device: {param.device}, dtype: {param.dtype}"
                )

    def _load_fp16_state_dict_fn(self):
        if self.scaler is not None:
            return self.scaler.load_state_dict
        return None

    def barrier(self):
        self._runner.barrier()

    # ------------------------------------------------------------- properties
    @property
    def step_loss(self):
        return self._coerce_loss(self._last_step_loss)

    @staticmethod
    def _coerce_loss(val):
        """Public properties expose plain floats; internal tracking may hold
        lazy (async-synced) values."""
        if isinstance(val, (list, tuple)):
            return type(val)(float(v) for v in val)
        return float(val)

    @property
    def model_access(self):
        """The bare user module regardless of engine wrapping."""
        if isinstance(self._model, (StokeDDPModule, StokeSDDPModule, StokeFSDPModule)):
            return self._model.module
        return self._model

    @property
    def loss_access(self):
        return self._loss

    @property
    def optimizer(self):
        return self._optimizer

    @property
    def scaler(self):
        return self._runner.scaler

    @property
    def fp16_state_dict(self):
        if self.scaler is not None:
            return self.scaler.state_dict()
        return None

    @property
    def status(self):
        return self._status

    @property
    def batch_size(self):
        return self._status.batch_size

    @property
    def effective_batch_size(self):
        return self._status.effective_batch_size

    @property
    def grad_clip(self):
        return self._status.grad_clip

    @property
    def grad_accum(self):
        return self._status.grad_accum

    @property
    def gpu(self):
        return self._status.gpu

    @property
    def cuda(self):
        return self._status.cuda

    @property
    def nccl(self):
        return self._status.nccl

    @property
    def fp16(self):
        return self._status.fp16

    @property
    def is_apex(self):
        return self._status.is_fp16_apex

    @property
    def is_amp(self):
        return self._status.is_fp16_amp

    @property
    def distributed(self):
        return self._status.distributed

    @property
    def is_ddp(self):
        return self._status.is_distributed_ddp

    @property
    def is_horovod(self):
        return self._status.is_distributed_horovod

    @property
    def is_deepspeed(self):
        return self._status.is_distributed_deepspeed

    @property
    def oss(self):
        return self._status.oss

    @property
    def sharded(self):
        return self._status.sharded

    @property
    def fully_sharded(self):
        return self._status.fully_sharded

    @property
    def world_size(self):
        return self._runner.world_size

    @property
    def rank(self):
        return self._runner.rank

    @property
    def amp_config(self):
        return self._status.amp_config if self.is_amp else None

    @property
    def apex_config(self):
        return self._status.apex_config if self.is_apex else None

    @property
    def ddp_config(self):
        return self._status.ddp_config if self.is_ddp else None

    @property
    def deepspeed_config(self):
        return self._status.deepspeed_config if self.is_deepspeed else None

    @property
    def oss_config(self):
        return self._status.oss_config if self.oss else None

    @property
    def sddp_config(self):
        return self._status.sddp_config if self.sharded else None

    @property
    def fsdp_config(self):
        return self._status.fsdp_config if self.fully_sharded else None

    @property
    def horovod_config(self):
        return self._status.horovod_config if self.is_horovod else None

    @property
    def num_model_parameters(self):
        return sum(p.numel() for p in self.model_access.parameters() if p.requires_grad)

    @property
    def ema_loss(self):
        return self._coerce_loss(self._rolling_mean_loss)
