from stoke.runtime.runner import StokeRunner  # noqa: F401
