# Direct Python access to the CDNA4 init kernels (tdx:: ops). The
# deferred-init replay engine calls these through the dispatcher redirect;
# this module is the public face for explicit use (e.g. custom
# materialization pipelines, tests, benchmarks).
#
# On non-GPU tensors (or when the _K extension is absent) every function
# falls back to the stock ATen op, so call sites stay portable.

from typing import Optional

import torch


def _native(t: torch.Tensor) -> bool:
    if not t.is_cuda or not t.is_contiguous():
        return False
    try:
        torch.ops.tdx.uniform_
        return True
    except (RuntimeError, AttributeError):
        return False


_RNG_DTYPES = (torch.float32, torch.bfloat16, torch.float16)


def uniform_(
    tensor: torch.Tensor,
    from_: float = 0.0,
    to: float = 1.0,
    *,
    seed: Optional[int] = None,
    offset: Optional[int] = None,
) -> torch.Tensor:
    """U(from_, to) fill. With explicit (seed, offset) the result is a pure
    function of those values — any slice, any device, any time (the
    counter-based Philox property sharded materialization relies on)."""
    if _native(tensor) and tensor.dtype in _RNG_DTYPES:
        return torch.ops.tdx.uniform_(tensor, from_, to, seed=seed, offset=offset)
    if seed is not None:
        g = torch.Generator(device=tensor.device)
        g.manual_seed(seed + (offset or 0))
        return tensor.uniform_(from_, to, generator=g)
    return tensor.uniform_(from_, to)


def normal_(
    tensor: torch.Tensor,
    mean: float = 0.0,
    std: float = 1.0,
    *,
    seed: Optional[int] = None,
    offset: Optional[int] = None,
) -> torch.Tensor:
    """N(mean, std) fill (Philox + Box-Muller on GPU)."""
    if _native(tensor) and tensor.dtype in _RNG_DTYPES:
        return torch.ops.tdx.normal_(tensor, mean, std, seed=seed, offset=offset)
    if seed is not None:
        g = torch.Generator(device=tensor.device)
        g.manual_seed(seed + (offset or 0))
        return tensor.normal_(mean, std, generator=g)
    return tensor.normal_(mean, std)


def fill_(tensor: torch.Tensor, value: float) -> torch.Tensor:
    if _native(tensor) and tensor.dtype in _RNG_DTYPES:
        return torch.ops.tdx.fill_(tensor, value)
    return tensor.fill_(value)


def zero_(tensor: torch.Tensor) -> torch.Tensor:
    if _native(tensor):
        return torch.ops.tdx.zero_(tensor)
    return tensor.zero_()


def bernoulli_(
    tensor: torch.Tensor,
    p: float = 0.5,
    *,
    seed: Optional[int] = None,
    offset: Optional[int] = None,
) -> torch.Tensor:
    """Bernoulli(p) 0/1 fill (uniform-threshold over the Philox stream)."""
    if _native(tensor) and tensor.dtype in _RNG_DTYPES:
        return torch.ops.tdx.bernoulli_(tensor, p, seed=seed, offset=offset)
    if seed is not None:
        g = torch.Generator(device=tensor.device)
        g.manual_seed(seed + (offset or 0))
        return tensor.bernoulli_(p, generator=g)
    return tensor.bernoulli_(p)
