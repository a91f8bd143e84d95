from stoke.nn.act_ckpt import apply_activation_checkpointing  # noqa: F401
from stoke.nn.fused_bn import FusedBNAct2d  # noqa: F401
from stoke.nn.layernorm import StokeLayerNorm  # noqa: F401
from stoke.nn.rmsnorm import StokeRMSNorm  # noqa: F401
from stoke.nn.fp8 import FP8Linear, convert_linears_to_fp8, fp8_available  # noqa: F401
from stoke.nn.swiglu import swiglu  # noqa: F401
from stoke.nn.cross_entropy import fused_cross_entropy  # noqa: F401
from stoke.nn.attention import attention, flash_attention  # noqa: F401
from stoke.nn.fp8_delayed import FP8LinearDelayed  # noqa: F401
