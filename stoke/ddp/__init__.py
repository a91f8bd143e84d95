from stoke.ddp.engine import StokeDDPModule  # noqa: F401
