from torchdistx_amd.ops.init_ops import (  # noqa: F401
    bernoulli_,
    fill_,
    normal_,
    uniform_,
    zero_,
)
