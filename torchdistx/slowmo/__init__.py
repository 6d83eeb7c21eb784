from torchdistx.slowmo import slowmo_comm, slowmo_optimizer  # noqa: F401
from torchdistx_amd.slowmo import (  # noqa: F401
    SlowMomentumOptimizer,
    SlowMoState,
    slowmo_hook,
)
