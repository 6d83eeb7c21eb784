# FSDP2 (fully_shard / DTensor) integration: deferred-init modules are
# materialized unit-by-unit and sharded, matching an eagerly-constructed
# fully_shard model bitwise. DTensor is a wrapper subclass and therefore
# cannot be recorded on the tape directly (same limitation as the
# reference framework); fully_shard_deferred is the supported flow.

import os

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from tests._dist_utils import run_distributed
from torchdistx_amd import deferred_init
from torchdistx_amd.deferred_init import is_deferred
from torchdistx_amd.parallel import fully_shard_deferred


def _make():
    torch.manual_seed(7)
    return nn.Sequential(nn.Linear(16, 16), nn.ReLU(), nn.Linear(16, 4))


@pytest.fixture
def single_proc_group():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29580")
    dist.init_process_group("gloo", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def test_fully_shard_deferred_single_rank(single_proc_group) -> None:
    # Pin the mesh to CPU: on a GPU box fully_shard's default mesh would
    # otherwise target cuda while this test's tensors live on CPU.
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import DTensor

    mesh = init_device_mesh("cpu", (1,))
    m = deferred_init(_make)
    assert is_deferred(m)
    fully_shard_deferred(m, submodules=[m[0], m[2]], mesh=mesh)
    assert not is_deferred(m)
    for p in m.parameters():
        assert isinstance(p.data, DTensor)

    x = torch.randn(3, 16)
    y = m(x)
    y.sum().backward()
    assert m[0].weight.grad is not None

    ref = _make()
    assert torch.equal(y.detach(), ref(x).detach())


def _fsdp2_worker(rank, world_size):
    from torch.distributed.device_mesh import init_device_mesh

    mesh = init_device_mesh("cpu", (world_size,))
    torch.manual_seed(rank)  # rank-skewed; deferred tape must override
    m = deferred_init(_make)
    fully_shard_deferred(m, submodules=[m[0], m[2]], mesh=mesh)

    torch.manual_seed(99)
    x = torch.randn(3, 16)  # same batch on every rank
    y = m(x)
    y.sum().backward()

    ref = _make()
    return (
        torch.equal(y.detach(), ref(x).detach()),
        m[0].weight.grad is not None,
    )


def test_fully_shard_deferred_two_ranks() -> None:
    results = run_distributed(_fsdp2_worker, world_size=2)
    for matches_eager, has_grad in results:
        assert matches_eager
        assert has_grad
