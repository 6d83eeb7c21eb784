from torchdistx_amd.optimizers import AnyPrecisionAdamW  # noqa: F401
