from torchdistx_amd.optimizers.anyprecision_optimizer import (  # noqa: F401
    AnyPrecisionAdamW,
)
