"""FSDP2 (``fully_shard``) integration for deferred-init modules.

``fully_shard`` cannot consume a still-deferred module directly: DTensor
is a tensor *wrapper subclass*, so the chunk/pad/copy ops it issues on
the inner local tensors dispatch from inside another handler's frame and
can never reach the deferred-init recorder (the reference framework has
the same limitation — wrapper subclasses are outside the tape's op
model). The supported composition is therefore *materialize, then
shard* — done one FSDP unit at a time so peak memory is bounded by the
largest unit's full parameters, not the whole model:

    model = deferred_init(build, cfg, device="cuda")
    fully_shard_deferred(model, submodules=model.blocks)

After the call every parameter is a regular FSDP2 ``DTensor`` sharded
parameter, bitwise-identical to eagerly constructing the module with the
same seed and calling ``fully_shard`` on it.

For a zero-redundancy init of models that do not fit one rank even
transiently per-unit, use :func:`materialize_module_dim0_sharded` /
:func:`materialize_module_dtensor` (slice materialization) instead.
"""

from typing import Iterable, Optional

import torch.nn as nn

from torchdistx_amd.deferred_init import is_deferred, materialize_module

__all__ = ["fully_shard_deferred"]


def fully_shard_deferred(
    module: nn.Module,
    *,
    submodules: Optional[Iterable[nn.Module]] = None,
    **fully_shard_kwargs,
) -> nn.Module:
    """Materializes a deferred-init ``module`` and applies FSDP2
    ``fully_shard``, unit by unit.

    Args:
        module:
            The root module returned by ``deferred_init``. Already
            materialized (or eagerly built) modules are accepted and
            just sharded.
        submodules:
            The FSDP units (e.g. ``model.blocks``); each is
            materialized then sharded in order, so at most one unit's
            full parameters are ever resident beyond their shards. The
            root is always processed last (catching parameters outside
            every unit, e.g. embeddings and the final norm).
        fully_shard_kwargs:
            Forwarded to ``torch.distributed.fsdp.fully_shard``
            (``mesh=…``, ``reshard_after_forward=…``, ...).
    """
    from torch.distributed.fsdp import fully_shard

    units = list(submodules) if submodules is not None else []
    for sub in units:
        if is_deferred(sub):
            materialize_module(sub)
        fully_shard(sub, **fully_shard_kwargs)
    if is_deferred(module):
        materialize_module(module)
    fully_shard(module, **fully_shard_kwargs)
    return module
