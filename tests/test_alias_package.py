# The `torchdistx` alias package must behave as a drop-in for the
# reference: submodule layout, public names, and torch FSDP's built-in
# detection (torch/distributed/fsdp/_init_utils.py:55).

import torch


def test_alias_submodules_and_api() -> None:
    import torchdistx  # noqa: F401  (import is the point)
    from torchdistx import deferred_init, fake

    m = deferred_init.deferred_init(torch.nn.Linear, 4, 4)
    assert fake.is_fake(m.weight)
    assert deferred_init.is_deferred(m)
    deferred_init.materialize_module(m)
    assert not deferred_init.is_deferred(m)

    from torchdistx.gossip_grad import GossipGraDState, Topology  # noqa: F401
    from torchdistx.optimizers import AnyPrecisionAdamW  # noqa: F401
    from torchdistx.slowmo import (  # noqa: F401
        SlowMomentumOptimizer,
        SlowMoState,
        slowmo_hook,
    )


def test_fsdp_detects_torchdistx() -> None:
    import importlib

    import torch.distributed.fsdp._init_utils as init_utils

    importlib.reload(init_utils)
    assert init_utils._TORCHDISTX_AVAIL
