from stoke.nn.fused_bn import FusedBNAct2d  # noqa: F401
