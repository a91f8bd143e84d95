from stoke.amp.scaler import StokeGradScaler, StokePerLossScaler  # noqa: F401
