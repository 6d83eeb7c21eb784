# FSDP integration on GPU: the deferred_init -> FSDP(param_init_fn=...)
# flow the feature exists for, plus the comm-hook registration path.
# Multi-rank hook numerics are covered CPU-side in test_comm_hooks_multiproc
# and test_gossip_grad (gloo); tests here follow the reference's
# skip_if_lt_x_gpu pattern for >1-GPU cases.

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture()
def nccl_world1():
    if not torch.cuda.is_available():
        pytest.skip("needs a ROCm GPU")
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    import socket

    with socket.socket() as _s:
        _s.bind(("127.0.0.1", 0))
        os.environ["MASTER_PORT"] = str(_s.getsockname()[1])
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    yield dist
    dist.destroy_process_group()


@pytest.mark.parametrize("explicit_init_fn", [True, False])
def test_fsdp_with_deferred_init(nccl_world1, explicit_init_fn) -> None:
    # Without param_init_fn, FSDP's built-in torchdistx support (enabled by
    # the `torchdistx` alias package) detects the fake parameters and calls
    # materialize_module itself.
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

    from torchdistx_amd import deferred_init, materialize_module
    from torchdistx_amd.models import TINY, build_model

    torch.manual_seed(0)
    module = deferred_init(
        build_model, TINY, device="cuda", dtype=torch.float32
    )
    kwargs = {}
    if explicit_init_fn:
        kwargs["param_init_fn"] = lambda m: materialize_module(m)
    fsdp = FSDP(
        module,
        device_id=torch.cuda.current_device(),
        **kwargs,
    )
    tokens = torch.randint(0, TINY.vocab_size, (2, 16), device="cuda")
    loss = fsdp(tokens[:, :-1]).float().mean()
    loss.backward()
    torch.cuda.synchronize()
    assert loss.isfinite().item()


def test_slowmo_hook_registers_on_fsdp(nccl_world1) -> None:
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP
    from torch.distributed.fsdp import ShardingStrategy

    from torchdistx_amd.slowmo import (
        SlowMomentumOptimizer,
        SlowMoState,
        slowmo_hook,
    )

    dist = nccl_world1
    net = torch.nn.Linear(8, 8).cuda()
    fsdp = FSDP(net, sharding_strategy=ShardingStrategy.NO_SHARD)
    state = SlowMoState(dist.group.WORLD, sync_grads=True)
    fsdp.register_comm_hook(state, slowmo_hook)

    optim = SlowMomentumOptimizer(
        torch.optim.SGD(fsdp.parameters(), lr=0.1), slowmo_freq=2
    )
    for _ in range(3):
        optim.zero_grad()
        fsdp(torch.randn(4, 8, device="cuda")).square().mean().backward()
        optim.step()
    torch.cuda.synchronize()
    sd = optim.state_dict()
    assert sd["step"] == 3


def test_get_num_modules_counts_nested_fsdp(nccl_world1) -> None:
    # Reference parity: get_num_modules counts nested FSDP units including
    # the root (reference gossip_grad.py:319-331, tested at
    # test_comm_hooks_fsdp.py:641-651).
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

    from torchdistx_amd.gossip_grad import get_num_modules

    inner = FSDP(torch.nn.Linear(4, 4).cuda())
    outer = FSDP(
        torch.nn.Sequential(inner, torch.nn.Linear(4, 4).cuda())
    )
    assert get_num_modules(outer) == 2
    assert get_num_modules(inner) == 1


def test_fsdp2_fully_shard_deferred(nccl_world1) -> None:
    # FSDP2: deferred_init -> fully_shard_deferred materializes unit-by-
    # unit through the CDNA4 init kernels and shards into DTensors.
    from torch.distributed.tensor import DTensor

    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import is_deferred
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import fully_shard_deferred

    torch.manual_seed(0)
    module = deferred_init(build_model, TINY, device="cuda",
                           dtype=torch.float32)
    assert is_deferred(module)
    fully_shard_deferred(module, submodules=list(module.blocks))
    assert not is_deferred(module)
    for p in module.parameters():
        assert isinstance(p.data, DTensor)

    torch.manual_seed(1)
    x = torch.randint(0, TINY.vocab_size, (2, 16), device="cuda")
    out = module(x)
    out.float().square().mean().backward()
    assert next(module.parameters()).grad is not None

    # Bitwise parity with the plain deferred_init -> materialize_module
    # flow (same tape, same CDNA4 kernels, no sharding).
    from torchdistx_amd import materialize_module

    torch.manual_seed(0)
    ref = deferred_init(build_model, TINY, device="cuda",
                        dtype=torch.float32)
    materialize_module(ref)
    torch.manual_seed(1)
    xr = torch.randint(0, TINY.vocab_size, (2, 16), device="cuda")
    assert torch.equal(out.detach(), ref(xr).detach())
    torch.cuda.synchronize()


def test_fsdp2_anyprecision_optimizer_step(nccl_world1) -> None:
    # FSDP2 DTensor parameters unwrap to their plain contiguous local
    # shard for the fused HIP AdamW step — the elementwise update on the
    # shard IS the sharded update — so the 3.4x single-kernel path must
    # actually run (asserted via the _fused_steps counter).
    from torchdistx_amd import deferred_init
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.optimizers import AnyPrecisionAdamW
    from torchdistx_amd.parallel import fully_shard_deferred

    torch.manual_seed(0)
    module = deferred_init(build_model, TINY, device="cuda",
                           dtype=torch.bfloat16)
    fully_shard_deferred(module, submodules=list(module.blocks))
    optim = AnyPrecisionAdamW(
        module.parameters(), lr=1e-3, use_kahan_summation=True
    )
    for step in range(3):
        x = torch.randint(0, TINY.vocab_size, (2, 16), device="cuda")
        loss = module(x).float().square().mean()
        optim.zero_grad()
        loss.backward()
        optim.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss.detach()).item()
    n_params = sum(1 for p in module.parameters() if p.grad is not None)
    assert optim._fused_steps >= 3 * max(n_params - 1, 1), (
        optim._fused_steps, n_params)


def test_shard_mode_batched_fill_bitwise(nccl_world1) -> None:
    # Shard-mode owned tensors fill through the batched planner; at world
    # 1 this rank owns everything and must match plain materialization
    # bitwise.
    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_distributed

    torch.manual_seed(17)
    ref = deferred_init(build_model, TINY, device="cuda",
                        dtype=torch.bfloat16)
    materialize_module(ref)

    torch.manual_seed(17)
    m = deferred_init(build_model, TINY, device="cuda",
                      dtype=torch.bfloat16)
    owner_map = materialize_module_distributed(m, mode="shard")
    assert owner_map and all(o == 0 for o in owner_map.values())
    torch.cuda.synchronize()
    for (n1, p1), (n2, p2) in zip(
        ref.named_parameters(), m.named_parameters()
    ):
        assert n1 == n2 and torch.equal(p1, p2), n1


def test_broadcast_and_allgather_modes_world1(nccl_world1) -> None:
    # World-1 exercises the full bucketed-broadcast machinery (bucket
    # planning, side-stream pipeline, RCCL broadcast-to-self, unpack
    # bypass on the owner) and the allgather mode on real RCCL; results
    # must match plain materialization bitwise.
    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_distributed

    for mode in ("broadcast", "allgather"):
        torch.manual_seed(23)
        ref = deferred_init(build_model, TINY, device="cuda",
                            dtype=torch.bfloat16)
        materialize_module(ref)

        torch.manual_seed(23)
        m = deferred_init(build_model, TINY, device="cuda",
                          dtype=torch.bfloat16)
        materialize_module_distributed(m, mode=mode)
        torch.cuda.synchronize()
        for (n1, p1), (n2, p2) in zip(
            ref.named_parameters(), m.named_parameters()
        ):
            assert n1 == n2 and torch.equal(p1, p2), (mode, n1)
