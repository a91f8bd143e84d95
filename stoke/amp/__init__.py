from stoke.amp.scaler import StokeGradScaler  # noqa: F401
