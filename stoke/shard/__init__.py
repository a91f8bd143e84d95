from stoke.shard.oss import OSSOptimizer  # noqa: F401
from stoke.shard.sddp import StokeSDDPModule  # noqa: F401
from stoke.shard.fsdp import StokeFSDPModule  # noqa: F401
