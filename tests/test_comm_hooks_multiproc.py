# Multi-process (gloo) tests for the SlowMo communication path and the
# distributed SlowMomentumOptimizer, mirroring the semantics the reference
# verifies under FSDP on GPUs (tests/python/test_comm_hooks_fsdp.py:110-260)
# on a CPU rig. FSDP-integration variants are in test_fsdp_gpu.py (gpu).

import pytest
import torch
import torch.distributed as dist

from tests._dist_utils import run_distributed
from torchdistx_amd.slowmo import SlowMomentumOptimizer, SlowMoState, slowmo_hook


def _slowmo_hook_sync(rank, world):
    state = SlowMoState(dist.group.WORLD, sync_grads=True)
    grad = torch.full([8], float(rank + 1))
    slowmo_hook(state, grad)
    return grad.tolist()


def test_slowmo_hook_averages_gradients() -> None:
    world = 2
    results = run_distributed(_slowmo_hook_sync, world)
    expected = [(1 + 2) / 2] * 8
    for got in results:
        assert got == pytest.approx(expected)


def _slowmo_hook_nosync(rank, world):
    state = SlowMoState(dist.group.WORLD, sync_grads=False)
    grad = torch.full([8], float(rank + 1))
    slowmo_hook(state, grad)
    return grad.tolist()


def test_slowmo_hook_no_sync_leaves_gradients() -> None:
    results = run_distributed(_slowmo_hook_nosync, 2)
    for rank, got in enumerate(results):
        assert got == pytest.approx([float(rank + 1)] * 8)


def _slowmo_subgroup_sync(rank, world):
    # world=4 as two 2-rank "nodes": averaging stays inside the node.
    subgroup, _ = dist.new_subgroups(group_size=2)
    state = SlowMoState(subgroup, sync_grads=True)
    grad = torch.full([4], float(rank))
    slowmo_hook(state, grad)
    return grad.tolist()


def test_slowmo_hook_subgroup_scope() -> None:
    results = run_distributed(_slowmo_subgroup_sync, 4)
    assert results[0] == results[1] == pytest.approx([0.5] * 4)
    assert results[2] == results[3] == pytest.approx([2.5] * 4)


def _slowmo_param_averaging(rank, world):
    # Different parameter values per rank; after slowmo_freq+1 steps with
    # zero gradients the periodic averager must have synchronized them.
    model = torch.nn.Linear(2, 2, bias=False)
    with torch.no_grad():
        model.weight.fill_(float(rank))
    opt = SlowMomentumOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1),
        slowmo_freq=2,
        slowmo_factor=0.0,
        slowmo_lr=1.0,
    )
    for _ in range(3):
        for p in model.parameters():
            p.grad = torch.zeros_like(p)
        opt.step()
    return model.weight.detach().flatten().tolist()


def test_slowmo_optimizer_averages_parameters() -> None:
    world = 2
    results = run_distributed(_slowmo_param_averaging, world)
    expected = [0.5] * 4  # average of fill values 0 and 1
    for got in results:
        assert got == pytest.approx(expected)
