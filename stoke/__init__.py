# -*- coding: utf-8 -*-
"""stoke-amd: an MI355X-native declarative training wrapper for PyTorch-ROCm.

Drop-in public surface of fidelity/stoke (same ``__all__`` as reference
``stoke/__init__.py:17-43``) rebuilt from scratch on a single
RCCL-over-xGMI process group, in-house DDP/OSS/SDDP/FSDP engines and
hand-written HIP/CDNA4 kernels.
"""

from stoke.configs import (
    AMPConfig,
    ApexConfig,
    BackendOptions,
    ClipGradConfig,
    ClipGradNormConfig,
    DDPConfig,
    DeepspeedAIOConfig,
    DeepspeedActivationCheckpointingConfig,
    DeepspeedConfig,
    DeepspeedFP16Config,
    DeepspeedFlopsConfig,
    DeepspeedOffloadOptimizerConfig,
    DeepspeedOffloadParamConfig,
    DeepspeedPLDConfig,
    DeepspeedTensorboardConfig,
    DeepspeedZeROConfig,
    FairscaleFSDPConfig,
    FairscaleOSSConfig,
    FairscaleSDDPConfig,
    HorovodConfig,
    HorovodOps,
    OffloadDevice,
    StokeOptimizer,
)
from stoke.data import BucketedDistributedSampler
from stoke.status import DistributedOptions, FP16Options
from stoke.stoke import Stoke
from stoke.utils import ParamNormalize

__all__ = [
    "Stoke",
    "ParamNormalize",
    "FP16Options",
    "DistributedOptions",
    "StokeOptimizer",
    "ClipGradNormConfig",
    "ClipGradConfig",
    "FairscaleOSSConfig",
    "FairscaleSDDPConfig",
    "FairscaleFSDPConfig",
    "HorovodConfig",
    "ApexConfig",
    "DeepspeedConfig",
    "DDPConfig",
    "AMPConfig",
    "DeepspeedAIOConfig",
    "DeepspeedActivationCheckpointingConfig",
    "DeepspeedFlopsConfig",
    "DeepspeedFP16Config",
    "DeepspeedPLDConfig",
    "DeepspeedOffloadOptimizerConfig",
    "DeepspeedOffloadParamConfig",
    "DeepspeedTensorboardConfig",
    "DeepspeedZeROConfig",
    "BucketedDistributedSampler",
]

from stoke._version import get_versions as _get_versions

__version__ = _get_versions()["version"]
del _get_versions
