# SlowMo tests. Coverage model: reference
# tests/python/test_comm_hooks_fsdp.py:210-407 (closed-form momentum check,
# checkpoint round-trip + failure cases, constructor validation,
# _prev_parameters bookkeeping, add_param_group) re-hosted on a CPU/gloo
# rig so they run without GPUs; the hook's all-reduce semantics are covered
# in test_comm_hooks_multiproc.py.

import os

import pytest
import torch
import torch.distributed as dist

from torchdistx_amd.slowmo import SlowMomentumOptimizer


@pytest.fixture()
def single_proc_group():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    import socket

    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        os.environ["MASTER_PORT"] = str(_s.getsockname()[1])
    dist.init_process_group("gloo", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def _model():
    torch.manual_seed(0)
    return torch.nn.Linear(4, 3)


def test_ctor_validation(single_proc_group) -> None:
    model = _model()
    base = torch.optim.SGD(model.parameters(), lr=0.1)

    with pytest.raises(ValueError, match="required"):
        SlowMomentumOptimizer(None)  # type: ignore[arg-type]
    with pytest.raises(ValueError, match="slowmo_freq"):
        SlowMomentumOptimizer(base, slowmo_freq=0)
    with pytest.raises(ValueError, match="slowmo_factor"):
        SlowMomentumOptimizer(base, slowmo_factor=-1.0)
    with pytest.raises(ValueError, match="slowmo_lr"):
        SlowMomentumOptimizer(base, slowmo_lr=-0.5)


def test_momentum_closed_form(single_proc_group) -> None:
    # With world size 1 the averaging stage is a no-op. The first
    # slow-momentum round fires on call freq+1 (the averager counts from 0
    # and the momentum check follows its post-increment step), and computes
    #   m    = (prev - p_after_sgd) / lr        (initial momentum is 0)
    #   prev = prev - slowmo_lr * lr * m
    #   p    = prev
    lr, slowmo_lr, freq = 0.1, 0.7, 3
    model = _model()
    base = torch.optim.SGD(model.parameters(), lr=lr)
    opt = SlowMomentumOptimizer(
        base, slowmo_freq=freq, slowmo_factor=0.5, slowmo_lr=slowmo_lr
    )

    init = [p.detach().clone() for p in model.parameters()]

    n_steps = freq + 1
    grads = []
    for step in range(n_steps):
        gs = []
        for p in model.parameters():
            g = torch.full_like(p, 0.01 * (step + 1))
            p.grad = g.clone()
            gs.append(g)
        grads.append(gs)
        opt.step()

    for i, p in enumerate(model.parameters()):
        p_sgd = init[i].clone()
        for step in range(n_steps):
            p_sgd -= lr * grads[step][i]
        momentum = (init[i] - p_sgd) / lr
        expected = init[i] - slowmo_lr * lr * momentum
        assert torch.allclose(p, expected, atol=1e-6), i


def test_state_dict_roundtrip(single_proc_group) -> None:
    model = _model()
    opt = SlowMomentumOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1),
        slowmo_freq=5,
        slowmo_factor=0.4,
        slowmo_lr=0.9,
    )
    for _ in range(3):
        for p in model.parameters():
            p.grad = torch.ones_like(p)
        opt.step()

    sd = opt.state_dict()
    assert sd["slowmo_freq"] == 5
    assert sd["slowmo_factor"] == 0.4
    assert sd["slowmo_lr"] == 0.9
    assert sd["step"] == 3

    model2 = _model()
    opt2 = SlowMomentumOptimizer(
        torch.optim.SGD(model2.parameters(), lr=0.1), slowmo_freq=2
    )
    opt2.load_state_dict(sd)
    assert opt2.slowmo_freq == 5
    assert opt2.averager.period == 5
    assert opt2.slowmo_factor == 0.4
    assert opt2.slowmo_lr == 0.9
    assert opt2.averager.step == 3


def test_load_state_dict_requires_lr(single_proc_group) -> None:
    model = _model()
    opt = SlowMomentumOptimizer(torch.optim.SGD(model.parameters(), lr=0.1))
    sd = opt.state_dict()
    del sd["param_groups"][0]["lr"]
    with pytest.raises((ValueError, KeyError)):
        opt.load_state_dict(sd)


def test_add_param_group_extends_prev_parameters(single_proc_group) -> None:
    model = _model()
    opt = SlowMomentumOptimizer(torch.optim.SGD(model.parameters(), lr=0.1))
    n_before = len(opt._prev_parameters)
    extra = torch.nn.Linear(2, 2)
    opt.add_param_group({"params": list(extra.parameters()), "lr": 0.05})
    assert len(opt._prev_parameters) == n_before + 2
    assert len(opt.param_groups) == 2


def test_base_optimizer_without_params_rejected(single_proc_group) -> None:
    with pytest.raises((ValueError, Exception)):
        SlowMomentumOptimizer(torch.optim.SGD([], lr=0.1))
