from torchdistx_amd.ops.init_ops import (  # noqa: F401
    fill_,
    normal_,
    uniform_,
    zero_,
)
