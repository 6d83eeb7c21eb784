# Alias of torchdistx_amd.deferred_init (see package __init__).
from torchdistx_amd.deferred_init import (  # noqa: F401
    deferred_init,
    is_deferred,
    materialize_module,
    materialize_tensor,
)
