# -*- coding: utf-8 -*-
"""Run-state validation and configuration resolution.

``StokeStatus`` holds the declared run state (flags + typed configs), runs the
invalid-combination rule matrix up front, and lazily resolves per-backend
configs with defaults.  The rule matrix mirrors the reference's checks
(``stoke/status.py:192-289``) so user-facing error behavior is identical, with
"CUDA"/"NCCL" meaning ROCm/RCCL availability through the torch APIs.
"""

import os
from enum import Enum
from typing import List, Optional, Union

import attr
import torch

from stoke.configs import (
    AMPConfig,
    ApexConfig,
    ClipGradConfig,
    ClipGradNormConfig,
    DDPConfig,
    DeepspeedConfig,
    DeepspeedFP16Config,
    FairscaleFSDPConfig,
    FairscaleOSSConfig,
    FairscaleSDDPConfig,
    HorovodConfig,
)


class DistributedOptions(Enum):
    """Distributed mode options.

    All three map onto the single RCCL-over-xGMI process group; they are kept
    as distinct options for reference API compatibility ("horovod" runs the
    same in-house DDP engine, "deepspeed" additionally maps ZeRO stages onto
    the in-house shard engine).
    """

    horovod = "horovod"
    ddp = "ddp"
    deepspeed = "deepspeed"


class FP16Options(Enum):
    """Mixed-precision mode options.

    "amp" is the native dynamic-loss-scale fp16 path; "apex_O1"/"apex_O2" and
    "deepspeed" are accepted for compatibility and run the same native path.
    """

    apex_O1 = "apex_O1"
    apex_O2 = "apex_O2"
    amp = "amp"
    deepspeed = "deepspeed"
    # Native extension: bf16 autocast needs no loss scaler on CDNA4 and is the
    # preferred mixed-precision mode on MI355X.
    bf16 = "bf16"


class _MissingLocalRankException(Exception):
    """Raised when a local rank cannot be resolved from config or env."""

    pass


# Internal evolved FSDP config carrying the runtime-derived mixed_precision flag
# (reference keeps this in extensions.py:25-27, injected at status.py:596-614).
@attr.s(auto_attribs=True)
class _FairscaleFSDPConfig(FairscaleFSDPConfig):
    mixed_precision: bool = False


class StokeStatus:
    """Validates and exposes the requested run configuration.

    Attributes are surfaced via properties; invalid flag combinations raise at
    construction time with the same error classes/messages style as the
    reference (``status.py:54-654``).
    """

    def __init__(
        self,
        batch_size_per_device: int,
        grad_accum: Optional[int],
        grad_clip: Optional[Union[ClipGradConfig, ClipGradNormConfig]],
        gpu: bool,
        fp16: Optional[FP16Options],
        distributed: Optional[DistributedOptions],
        fairscale_oss: bool,
        fairscale_sddp: bool,
        fairscale_fsdp: bool,
        configs: Optional[List] = None,
    ):
        self._key_list = [
            "AMPConfig",
            "ApexConfig",
            "DDPConfig",
            "DeepspeedConfig",
            "FairscaleOSSConfig",
            "FairscaleSDDPConfig",
            "FairscaleFSDPConfig",
            "HorovodConfig",
        ]
        self._configs = self._set_configs(configs=configs)
        if (grad_clip is not None) and not isinstance(
            grad_clip, (ClipGradConfig, ClipGradNormConfig)
        ):
            raise TypeError(
                "Stoke -- grad_clip argument must be of type ClipGradConfig or ClipGradNormConfig"
            )
        # Normalize enum values to raw strings so both the enum and the raw
        # string are accepted everywhere downstream
        distributed = getattr(distributed, "value", distributed)
        fp16 = getattr(fp16, "value", fp16)
        self._status = {
            "cuda": torch.cuda.is_available(),
            "nccl": torch.distributed.is_nccl_available(),
            "batch_size": batch_size_per_device,
            "grad_accum": grad_accum if grad_accum is not None else 1,
            "grad_clip": grad_clip,
            "gpu": gpu,
            "distributed": distributed,
            "zero": self._configs.get("DeepspeedConfig").zero_optimization.stage
            if self._configs.get("DeepspeedConfig")
            and self._configs.get("DeepspeedConfig").zero_optimization is not None
            else None,
            "oss": fairscale_oss,
            "sharded": fairscale_sddp,
            "fully_sharded": fairscale_fsdp,
            "world_size": -1,
        }
        self._status.update({"fp16": self._set_fp16(fp16=fp16)})
        self._check_all_raised_combinations()

    def _check_all_raised_combinations(self):
        """Raise on every invalid flag combination (reference rule matrix)."""
        # Rule 1: no GPU flag without a visible device
        if self.gpu and not self.cuda:
            raise ValueError("Stoke -- GPU(s) cannot be used as CUDA is not available")
        # Rule 2: fairscale-style sharding and deepspeed-style config are exclusive
        if self.is_fairscale and (
            self.is_distributed_deepspeed or self.is_fp16_deepspeed
        ):
            raise ValueError(
                f"Stoke -- Cannot use both fairscale extensions "
                f"(currently: oss: {self.oss}, sddp: {self.sharded}) "
                f"and deepspeed (currently: distributed: {self.is_distributed_deepspeed}, "
                f"fp16: {self.is_fp16_deepspeed})"
            )
        # Rule 3: distributed needs device + RCCL
        if (
            not self.cuda or not self.gpu or not self.nccl
        ) and self.distributed is not None:
            raise ValueError(
                f"Stoke -- Distributed requires CUDA (currently: {self.cuda}), GPU (currently: {self.gpu}), "
                f"and NCCL (currently: {self.nccl})"
            )
        # Rule 4: no mixed precision without a device
        if not self.cuda and (self.fp16 is not None):
            raise ValueError("Stoke -- FP16 training requires CUDA availability")
        # Rule 5: sharding requires distributed DDP mode on device
        if (
            not self.cuda
            or not self.gpu
            or not self.nccl
            or not self.is_distributed_ddp
        ) and self.is_fairscale:
            raise ValueError(
                f"Stoke -- Fairscale extensions (currently: oss: {self.oss}, sddp: {self.sharded}) "
                f"requires CUDA (currently: {self.cuda}), "
                f"GPU (currently: {self.gpu}), "
                f"DDP (currently: {self.is_distributed_ddp}) and NCCL (currently: {self.nccl})"
            )
        # Rule 6: SDDP (grad shard) needs OSS (optimizer shard)
        if self.sharded and not self.oss:
            raise ValueError(
                f"Stoke -- Fairscale SDDP requires OSS (currently: oss: {self.oss}, sddp: {self.sharded})"
            )
        # Rule 7: FSDP manages its own optimizer shard; exclusive with OSS/SDDP
        if (self.sharded or self.oss) and self.fully_sharded:
            raise ValueError(
                f"Stoke -- Fairscale FSDP does not require SDDP or OSS as it manages OSS itself"
                f"(currently: oss: {self.oss}, sddp: {self.sharded}. fsdp: {self.fully_sharded})"
            )
        # Rule 8: apex-style modes are not supported with sharding
        if self.is_fairscale and self.is_fp16_apex:
            raise ValueError(
                f"Stoke -- Fairscale does not currently support APEX (currently: {self.is_fp16_apex}) "
                f"for mixed precision"
            )
        # Rule 9: sharded optimizers cannot clip by value (owner-only grads)
        if (self.oss or self.fully_sharded) and isinstance(
            self.grad_clip, ClipGradConfig
        ):
            raise ValueError(
                f"Stoke -- Fairscale OSS and FSDP do not currently support torch.nn.utils.clip_grad_value_ "
                f"(currently: {type(self.grad_clip).__name__})"
            )
        # Rule 10: deepspeed fp16 requires deepspeed distributed
        if self.is_fp16_deepspeed and not self.is_distributed_deepspeed:
            raise ValueError(
                f"Stoke -- Deepspeed FP16 (currently: {self.is_fp16_deepspeed}) requires the use of "
                f"Deepspeed distributed (currently: {self.is_distributed_deepspeed})"
            )
        # Rule 11: deepspeed distributed only pairs with deepspeed fp16
        if (
            self.is_distributed_deepspeed
            and self.fp16 is not None
            and not self.is_fp16_deepspeed
        ):
            raise ValueError(
                f"Stoke -- Deepspeed distributed (currently: {self.is_distributed_deepspeed}) only "
                f"supports its own internal FP16 implementation (currently: {self.fp16})"
            )
        # Rule 12: ZeRO > 0 requires deepspeed fp16
        if (
            self.is_distributed_deepspeed
            and self.zero is not None
            and self.zero > 0
            and not self.is_fp16_deepspeed
        ):
            raise ValueError(
                f"Stoke -- Deepspeed ZeRO extension (currently: Stage-{self.zero}) requires Deepspeed"
                f"FP16 extension (currently: {self.is_fp16_deepspeed})"
            )

    def _set_fp16(self, fp16: Optional[str]):
        """Accept the fp16 mode only when a device is present (reference behavior)."""
        if self._status.get("cuda") and (fp16 is not None):
            return fp16
        return None

    def _set_configs(self, configs):
        """Key user configs by class name; missing keys resolve to None."""
        if configs is not None:
            config_dict = {type(val).__name__: val for val in configs}
        else:
            config_dict = {}
        none_dict = {val: None for val in self._key_list if val not in config_dict}
        config_dict.update(none_dict)
        return config_dict

    def set_post_init_values(self, world_size: int):
        """Record values only known after process-group init."""
        self._status.update({"world_size": world_size})

    # ------------------------------------------------------------------ state
    @property
    def status(self):
        return self._status

    @property
    def batch_size(self):
        return self._status.get("batch_size")

    @property
    def effective_batch_size(self):
        return self.batch_size * self.grad_accum * self._status.get("world_size")

    @property
    def grad_clip(self):
        return self._status.get("grad_clip")

    @property
    def grad_accum(self):
        return self._status.get("grad_accum")

    @property
    def gpu(self):
        return self._status.get("gpu")

    @property
    def cuda(self):
        return self._status.get("cuda")

    @property
    def nccl(self):
        return self._status.get("nccl")

    @property
    def fp16(self):
        return self._status.get("fp16")

    @property
    def is_fp16_apex(self):
        return self.fp16 == "apex_O1" or self.fp16 == "apex_O2"

    @property
    def is_fp16_amp(self):
        return self.fp16 == "amp"

    @property
    def is_fp16_bf16(self):
        return self.fp16 == "bf16"

    @property
    def is_fp16_deepspeed(self):
        return self.fp16 == "deepspeed"

    @property
    def oss(self):
        return self._status.get("oss")

    @property
    def sharded(self):
        return self._status.get("sharded")

    @property
    def fully_sharded(self):
        return self._status.get("fully_sharded")

    @property
    def world_size(self):
        return self._status.get("world_size")

    @property
    def zero(self):
        return self._status.get("zero")

    @property
    def is_fairscale(self):
        return self.oss or self.sharded or self.fully_sharded

    @property
    def distributed(self):
        return self._status.get("distributed")

    @property
    def is_distributed_deepspeed(self):
        return self.distributed == "deepspeed"

    @property
    def is_distributed_ddp(self):
        return self.distributed == "ddp"

    @property
    def is_distributed_horovod(self):
        return self.distributed == "horovod"

    # -------------------------------------------------------- resolved configs
    @property
    def apex_config(self):
        config = self._configs.get("ApexConfig")
        return config if config is not None else ApexConfig()

    @property
    def amp_config(self):
        config = self._configs.get("AMPConfig")
        return config if config is not None else AMPConfig()

    @property
    def ddp_config(self):
        """Resolve the DDP config, discovering LOCAL_RANK from env if needed."""
        config = self._configs.get("DDPConfig")
        if config is not None and config.local_rank is None:
            local_rank = self._env_local_rank()
            config = attr.evolve(config, local_rank=local_rank)
        elif config is None:
            config = DDPConfig(local_rank=self._env_local_rank())
        return config

    @staticmethod
    def _env_local_rank():
        try:
            return int(os.environ["LOCAL_RANK"])
        except KeyError:
            raise _MissingLocalRankException(
                "Stoke -- Device local rank must be defined within the DDPConfig "
                "(handled by parsing --local_arg from the torch.distributed.launch "
                "command) or defined as env variable LOCAL_RANK (handled by calling "
                "torch.distributed.launch with the --use_env flag)"
            )

    @property
    def deepspeed_config(self):
        config = self._configs.get("DeepspeedConfig")
        if self.fp16 == "deepspeed" and config is None:
            config = DeepspeedConfig(fp16=DeepspeedFP16Config())
        elif self.fp16 == "deepspeed" and config is not None and config.fp16 is None:
            config = attr.evolve(config, fp16=DeepspeedFP16Config())
        elif config is None:
            config = DeepspeedConfig()
        return config

    @property
    def oss_config(self):
        config = self._configs.get("FairscaleOSSConfig")
        return config if config is not None else FairscaleOSSConfig()

    @property
    def sddp_config(self):
        config = self._configs.get("FairscaleSDDPConfig")
        return config if config is not None else FairscaleSDDPConfig()

    @property
    def fsdp_config(self):
        """Resolve FSDP config with the runtime-derived mixed_precision flag."""
        config = self._configs.get("FairscaleFSDPConfig")
        if config is None:
            config = FairscaleFSDPConfig()
        config_dict = attr.asdict(config, recurse=False)
        config_dict.update({"mixed_precision": self.is_fp16_amp or self.is_fp16_bf16})
        return _FairscaleFSDPConfig(**config_dict)

    @property
    def horovod_config(self):
        config = self._configs.get("HorovodConfig")
        return config if config is not None else HorovodConfig()

    def __repr__(self):
        return (
            f"STOKE STATE:\n"
            f"    CUDA AVAILABLE: {self.cuda}\n"
            f"    NCCL AVAILABLE: {self.nccl}\n"
            f"    GPU FLAG: {self.gpu}\n"
            f"    FP16 FLAG: {self.fp16}\n"
            f"    DISTRIBUTED BACKEND: {self.distributed}\n"
            f"    FAIRSCALE OSS: {self.oss}\n"
            f"    FAIRSCALE SDDP: {self.sharded}\n"
            f"    FAIRSCALE FSDP: {self.fully_sharded}\n"
            f'    DEEPSPEED ZeRO: {f"Stage {self.zero}" if self.is_distributed_deepspeed else f"False"}\n'
            f"    WORLD SIZE: {self.world_size}\n"
            f"    GRAD ACCUMULATION STEPS: {self.grad_accum}\n"
            f"    BATCH SIZE (PER DEVICE): {self.batch_size}\n"
            f"    EFFECTIVE BATCH SIZE (ALL DEVICES): {self.effective_batch_size}\n"
            f'    GRAD CLIP: ({", ".join(f"{k}: {v}" for k, v in attr.asdict(self.grad_clip).items()) if self.grad_clip is not None else "None"})'
        )
