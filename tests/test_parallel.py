# Distributed materialization tests (gloo, CPU). GPU/RCCL variants live in
# the gpu-marked suite.

import pytest
import torch

from tests._dist_utils import run_distributed
from torchdistx_amd.parallel import assign_owners


def test_assign_owners_is_balanced_and_deterministic() -> None:
    sizes = [100, 90, 10, 10, 10, 60, 40]
    a = assign_owners(sizes, 2)
    b = assign_owners(sizes, 2)
    assert a == b
    loads = [0, 0]
    for s, o in zip(sizes, a):
        loads[o] += s
    assert abs(loads[0] - loads[1]) <= max(sizes)
    assert set(a) == {0, 1}


def _materialize_worker(rank, world, mode):
    from torchdistx_amd import deferred_init, _C
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_distributed

    torch.manual_seed(42)
    m = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
    owners = materialize_module_distributed(m, mode=mode)

    out = {}
    out["owners"] = owners
    out["fake"] = {
        name: bool(_C.is_fake(p))
        for name, p in list(m.named_parameters()) + list(m.named_buffers())
    }
    out["sums"] = {
        name: float(p.detach().double().sum())
        for name, p in list(m.named_parameters()) + list(m.named_buffers())
        if not _C.is_fake(p)
    }
    return out


def test_replicate_matches_local_materialize() -> None:
    results = run_distributed(_materialize_worker, 2, "replicate")

    # Local single-process reference: replicate mode replays the full tape
    # on every rank in the same order, so it matches a local materialize
    # exactly.
    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model

    torch.manual_seed(42)
    ref = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
    materialize_module(ref)
    ref_sums = {
        name: float(p.detach().double().sum())
        for name, p in list(ref.named_parameters()) + list(ref.named_buffers())
    }

    for r in results:
        assert not any(r["fake"].values())
        for name, s in ref_sums.items():
            assert s == pytest.approx(r["sums"][name], rel=1e-6), name


def test_broadcast_mode_is_rank_consistent() -> None:
    # On CPU the stock generator is sequential, so a partitioned replay
    # draws a different stream than a local replay — but every rank must end
    # with the SAME fully materialized model (owners broadcast their
    # tensors). Bitwise equality with local replay additionally holds on
    # GPU, where the pinned-Philox kernels make init partition-invariant
    # (covered by the gpu-marked tests).
    results = run_distributed(_materialize_worker, 2, "broadcast")
    for r in results:
        assert not any(r["fake"].values())
    assert results[0]["sums"] == results[1]["sums"]


def test_shard_mode_partitions_ownership() -> None:
    results = run_distributed(_materialize_worker, 2, "shard")
    owners = results[0]["owners"]
    assert owners == results[1]["owners"]
    assert set(owners.values()) == {0, 1}

    # Each rank materialized its own tensors and left the rest fake, and
    # every tensor is real on exactly one rank.
    names = list(results[0]["fake"].keys())
    for name in names:
        real_on = [r for r in range(2) if not results[r]["fake"][name]]
        assert len(real_on) == 1, name


def test_default_dtype_is_pinned_at_record_time() -> None:
    # Recording under bf16 default dtype then materializing after the
    # default has reverted must still produce bf16 parameters.
    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model

    m = deferred_init(build_model, TINY, device="cpu", dtype=torch.bfloat16)
    assert torch.get_default_dtype() == torch.float32
    materialize_module(m)
    assert m.tok_emb.weight.dtype == torch.bfloat16
    # Buffers created with an explicit fp32 dtype stay fp32.
    assert m.rope_cos.dtype == torch.float32


def _expert_shard_worker(rank, world):
    from torchdistx_amd import deferred_init, _C
    from torchdistx_amd.models import TINY_MOE, build_model
    from torchdistx_amd.parallel import materialize_experts_sharded

    torch.manual_seed(3)
    m = deferred_init(build_model, TINY_MOE, device="cpu", dtype=torch.float32)
    owners = materialize_experts_sharded(m)
    state = {}
    for name, p in m.named_parameters():
        state[name] = None if _C.is_fake(p) else float(p.detach().double().sum())
    return owners, state


def test_expert_sharded_materialization() -> None:
    results = run_distributed(_expert_shard_worker, 2)
    owners0, s0 = results[0]
    owners1, s1 = results[1]
    assert owners0 == owners1
    assert set(owners0.values()) == {0, 1}

    for name in s0:
        is_expert = ".experts." in name
        if not is_expert:
            # Shared parameters materialize on every rank (identically up
            # to the partition-invariant GPU path; on CPU both replay the
            # full shared subgraph from the same seed).
            assert s0[name] is not None and s1[name] is not None, name
        else:
            # Every expert parameter is real on exactly one rank.
            real_on = [r for r, s in enumerate((s0, s1)) if s[name] is not None]
            assert len(real_on) == 1, name


def test_slice_materialization_matches_full_native() -> None:
    # Slices of any tensor are bitwise-equal to the corresponding rows of a
    # full materialization through the native (pinned-Philox) path, at any
    # (even non-group-aligned) boundary.
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_tensor_shard

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(0)
        full = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        torch.manual_seed(0)
        part = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        materialize_module(full)

        w = full.tok_emb.weight.detach()
        s = [
            materialize_tensor_shard(part.tok_emb.weight, a, b)
            for a, b in ((0, 13), (13, 50), (50, 128))
        ]
        assert torch.equal(torch.cat(s), w)
        assert s[0].requires_grad == part.tok_emb.weight.requires_grad
    finally:
        _C.set_native_init_cpu(False)


def test_dim0_sharded_module_reconstructs_model() -> None:
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_dim0_sharded

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(5)
        full = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        materialize_module(full)
        reference = dict(full.named_parameters())

        world = 3
        gathered = {}
        for rank in range(world):
            torch.manual_seed(5)
            m = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
            for name, shard in materialize_module_dim0_sharded(
                m, rank=rank, world_size=world
            ).items():
                gathered.setdefault(name, []).append(shard)

        for name, ref in reference.items():
            got = torch.cat(gathered[name])
            assert torch.equal(got, ref.detach()), name
    finally:
        _C.set_native_init_cpu(False)


def test_slice_materialization_complex_tape_fallback() -> None:
    # A partial-tensor in-place op breaks the simple-chain property: the
    # C++ fast path must refuse it loudly, and the Python wrapper must fall
    # back to full materialization + slicing with correct values.
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_tensor_shard
    from torch.nn import Module, Parameter

    class M(Module):
        def __init__(self):
            super().__init__()
            a = torch.zeros(8, 8)
            v = a.view(-1)[:32].view(4, 8)
            v.add_(1)  # partial-tensor in-place: not a simple chain
            self.p = Parameter(a)

    m = deferred_init(M)
    with pytest.raises(RuntimeError, match="slice materialization"):
        _C.materialize_tensor_shard(m.p, 0, 4)

    shard = materialize_tensor_shard(m.p, 0, 4)
    expected = torch.ones(4, 8)
    assert torch.equal(shard.detach(), expected)
    tail = materialize_tensor_shard(m.p, 4, 8)
    assert torch.equal(tail.detach(), torch.zeros(4, 8))


def _allgather_worker(rank, world):
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_distributed

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(9)
        m = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        materialize_module_distributed(m, mode="allgather")
        torch.manual_seed(9)
        ref = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        materialize_module(ref)
        return all(
            torch.equal(a.detach(), b.detach())
            for (_, a), (_, b) in zip(m.named_parameters(), ref.named_parameters())
        )
    finally:
        _C.set_native_init_cpu(False)


def test_allgather_mode_reconstructs_native_full() -> None:
    # Per-rank slice init + all-gather equals a full native materialization
    # bitwise on every rank (uneven splits padded internally).
    assert all(run_distributed(_allgather_worker, 2))


def _dtensor_worker(rank, world):
    from torch.distributed.device_mesh import init_device_mesh

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model
    from torchdistx_amd.parallel import materialize_module_dtensor

    _C.set_native_init_cpu(True)
    try:
        mesh = init_device_mesh("cpu", (world,))
        torch.manual_seed(2)
        m = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        dts = materialize_module_dtensor(m, mesh)

        torch.manual_seed(2)
        ref = deferred_init(build_model, TINY, device="cpu", dtype=torch.float32)
        materialize_module(ref)
        refp = dict(list(ref.named_parameters()) + list(ref.named_buffers()))
        return all(
            torch.equal(dt.full_tensor(), refp[name].detach())
            for name, dt in dts.items()
        )
    finally:
        _C.set_native_init_cpu(False)


def test_dtensor_materialization_matches_full() -> None:
    # FSDP2-style: each rank materializes only its Shard(0) chunk straight
    # into DTensors; the assembled full tensors equal a native full
    # materialization bitwise.
    assert all(run_distributed(_dtensor_worker, 2))


def _dtensor_tp_worker(rank, world):
    from torch.distributed.device_mesh import init_device_mesh
    from torch.distributed.tensor import Replicate, Shard

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.parallel import materialize_module_dtensor

    class Block(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.up = torch.nn.Linear(16, 32)
            self.down = torch.nn.Linear(32, 16)

    shard_dims = {"down.weight": 1, "down.bias": None}

    _C.set_native_init_cpu(True)
    try:
        mesh = init_device_mesh("cpu", (world,))
        torch.manual_seed(12)
        m = deferred_init(Block)
        dts = materialize_module_dtensor(m, mesh, shard_dims=shard_dims)

        torch.manual_seed(12)
        ref = deferred_init(Block)
        materialize_module(ref)
        refp = dict(ref.named_parameters())
        ok = all(
            torch.equal(dt.full_tensor(), refp[name].detach())
            for name, dt in dts.items()
        )
        ok = ok and dts["down.weight"].placements == (Shard(1),)
        ok = ok and dts["down.bias"].placements == (Replicate(),)
        ok = ok and dts["up.weight"].placements == (Shard(0),)
        return ok
    finally:
        _C.set_native_init_cpu(False)


def test_dtensor_tp_placements_match_full() -> None:
    # shard_dims overrides per tensor: Shard(1) for row-parallel weights,
    # Replicate() (full bitwise copy, zero communication) for their
    # biases; everything still reassembles to the native full model.
    assert all(run_distributed(_dtensor_tp_worker, 2))


def test_slice_falls_back_for_explicit_generator() -> None:
    # An RNG op recorded with an explicit generator cannot use the
    # counter-based shard path; the wrapper transparently falls back to the
    # (generator-honoring) full replay and slices.
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_tensor_shard
    from torch.nn import Module, Parameter

    g = torch.Generator().manual_seed(3)

    class M(Module):
        def __init__(self):
            super().__init__()
            p = torch.empty(8, 4)
            p.uniform_(0, 1, generator=g)
            self.p = Parameter(p)

    m = deferred_init(M)
    with pytest.raises(RuntimeError, match="explicit generator"):
        _C.materialize_tensor_shard(m.p, 0, 4)
    shard = materialize_tensor_shard(m.p, 0, 4)

    g2 = torch.Generator().manual_seed(3)
    expected = torch.empty(8, 4).uniform_(0, 1, generator=g2)
    assert torch.equal(shard.detach(), expected[:4])


def test_slice_materialization_pointwise_chain() -> None:
    # trunc_normal_-style init records uniform_ -> erfinv_ -> mul_ ->
    # add_ -> clamp_; slicing commutes with the pointwise-scalar tail, so
    # the shard path must reproduce the full native materialization
    # bitwise on any row range.
    from torch.nn import Module, Parameter

    from torchdistx_amd import _C, deferred_init

    class M(Module):
        def __init__(self):
            super().__init__()
            w = torch.empty(64, 16)
            torch.nn.init.trunc_normal_(w, mean=0.1, std=0.7, a=-1.0, b=1.5)
            self.p = Parameter(w)

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(606)
        full_m = deferred_init(M)
        full = _C.materialize_tensor(full_m.p)

        torch.manual_seed(606)
        part_m = deferred_init(M)
        for start, end in [(0, 64), (5, 17), (63, 64), (0, 1), (32, 32)]:
            shard = _C.materialize_tensor_shard(part_m.p, start, end)
            assert torch.equal(shard, full.detach()[start:end]), (start, end)
    finally:
        _C.set_native_init_cpu(False)


def test_slice_materialization_any_dim_matches_full() -> None:
    # Any-dim slices (the windowed shard path, n_blocks > 1) are bitwise
    # sub-tensors of the full native materialization — group-aligned and
    # odd geometries, every dtype, every init kind.
    from torch.nn import Module, Parameter

    from torchdistx_amd import _C, deferred_init

    cases = [
        # (shape, dim, ranges) — (24, 40) is 8-aligned in the trailing
        # dim; (9, 11, 7) is odd everywhere (elementwise fallback).
        ((24, 40), 1, [(0, 40), (0, 8), (16, 40), (13, 27), (5, 5)]),
        ((9, 11, 7), 1, [(0, 11), (3, 8), (10, 11)]),
        ((9, 11, 7), 2, [(0, 7), (2, 5)]),
        ((6, 8, 16), 2, [(0, 16), (8, 16)]),
    ]
    for dtype in (torch.float32, torch.bfloat16, torch.float16):
        for init in ("normal", "uniform", "bernoulli", "trunc"):
            for shape, dim, ranges in cases:

                class M(Module):
                    def __init__(self):
                        super().__init__()
                        w = torch.empty(shape, dtype=dtype)
                        if init == "normal":
                            w.normal_(0.2, 1.3)
                        elif init == "uniform":
                            w.uniform_(-2.0, 3.0)
                        elif init == "bernoulli":
                            w.bernoulli_(0.4)
                        else:
                            torch.nn.init.trunc_normal_(w, std=0.5)
                        self.p = Parameter(w)

                _C.set_native_init_cpu(True)
                try:
                    torch.manual_seed(777)
                    full = _C.materialize_tensor(deferred_init(M).p).detach()
                    torch.manual_seed(777)
                    part = deferred_init(M)
                    for a, b in ranges:
                        shard = _C.materialize_tensor_shard(part.p, a, b, dim)
                        assert torch.equal(
                            shard, full.narrow(dim, a, b - a)
                        ), (dtype, init, shape, dim, a, b)
                finally:
                    _C.set_native_init_cpu(False)


def test_dim1_slices_reassemble_and_validate() -> None:
    # Concatenating every rank's dim-1 slice reconstructs the exact full
    # tensor (the row-parallel TP contract); bogus dims/ranges raise.
    import pytest

    from torch.nn import Module, Parameter

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_tensor_shard

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.empty(16, 48).normal_())

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(88)
        full = _C.materialize_tensor(deferred_init(M).p).detach()
        torch.manual_seed(88)
        part = deferred_init(M)
        world = 3
        cols = [
            materialize_tensor_shard(
                part.p, r * 48 // world, (r + 1) * 48 // world, dim=1
            )
            for r in range(world)
        ]
        assert torch.equal(torch.cat([c.detach() for c in cols], dim=1), full)
        assert cols[0].requires_grad

        with pytest.raises(RuntimeError, match="invalid slice dim"):
            _C.materialize_tensor_shard(part.p, 0, 4, 2)
        with pytest.raises(RuntimeError, match="invalid slice range"):
            _C.materialize_tensor_shard(part.p, 0, 49, 1)
    finally:
        _C.set_native_init_cpu(False)


def test_tp_sharded_module() -> None:
    # materialize_module_tp_sharded: listed tensors shard along their
    # assigned dim, unlisted ones replicate bitwise, unknown names raise.
    import pytest

    from torch.nn import Linear, Module

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_module_tp_sharded

    class Block(Module):
        def __init__(self):
            super().__init__()
            self.up = Linear(32, 64)    # column-parallel: shard dim 0
            self.down = Linear(64, 32)  # row-parallel: shard dim 1

    shard_dims = {"up.weight": 0, "up.bias": 0, "down.weight": 1}

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(99)
        full = deferred_init(Block)
        from torchdistx_amd.deferred_init import materialize_module

        materialize_module(full)
        reference = dict(full.named_parameters())

        world = 2
        gathered: dict = {}
        for rank in range(world):
            torch.manual_seed(99)
            m = deferred_init(Block)
            out = materialize_module_tp_sharded(
                m, shard_dims, rank=rank, world_size=world
            )
            for name, t in out.items():
                gathered.setdefault(name, []).append(t.detach())

        for name, parts in gathered.items():
            ref = reference[name].detach()
            dim = shard_dims.get(name)
            if dim is None:  # replicated (down.bias): identical full copies
                assert torch.equal(parts[0], parts[1]), name
                assert torch.equal(parts[0], ref), name
            else:
                assert torch.equal(torch.cat(parts, dim=dim), ref), name

        torch.manual_seed(99)
        m = deferred_init(Block)
        with pytest.raises(ValueError, match="not found in the module"):
            materialize_module_tp_sharded(m, {"nope.weight": 0})
        torch.manual_seed(99)
        m = deferred_init(Block)
        with pytest.raises(ValueError, match="out of range"):
            materialize_module_tp_sharded(m, {"up.bias": 1})
    finally:
        _C.set_native_init_cpu(False)


def test_shard_window_op_validation() -> None:
    # The windowed shard ops reject malformed windows loudly and no-op
    # on empty ones (the empty-per-rank-slice case).
    import pytest

    import torchdistx_amd  # noqa: F401  (registers tdx:: ops)

    t = torch.empty(6, dtype=torch.float32)
    with pytest.raises(RuntimeError, match="invalid shard window"):
        torch.ops.tdx.uniform_shard_win_(
            t, 2, 3, 2, 0, 0.0, 1.0, seed=1, offset=0
        )  # g_stride < block_len: blocks would overlap
    with pytest.raises(RuntimeError, match="n_blocks"):
        torch.ops.tdx.uniform_shard_win_(
            t, 2, 4, 8, 0, 0.0, 1.0, seed=1, offset=0
        )  # numel != n_blocks * block_len
    e = torch.empty(0, dtype=torch.float32)
    torch.ops.tdx.normal_shard_win_(e, 0, 0, 0, 0, 0.0, 1.0, seed=1, offset=0)
    assert e.numel() == 0


def test_slice_empty_ranges_any_dim() -> None:
    # world > dim size leaves some ranks with empty slices: shape must be
    # right and reassembly must still be exact, on both dims.
    from torch.nn import Module, Parameter

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_tensor_shard

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.empty(3, 5).normal_())

    _C.set_native_init_cpu(True)
    try:
        for dim in (0, 1):
            torch.manual_seed(17)
            full = _C.materialize_tensor(deferred_init(M).p).detach()
            torch.manual_seed(17)
            part = deferred_init(M)
            world = 7
            n = part.p.shape[dim]
            parts = []
            for r in range(world):
                a, b = r * n // world, (r + 1) * n // world
                s = materialize_tensor_shard(part.p, a, b, dim)
                assert s.shape[dim] == b - a
                parts.append(s.detach())
            assert torch.equal(torch.cat(parts, dim=dim), full)
    finally:
        _C.set_native_init_cpu(False)


def test_bernoulli_native_and_slice() -> None:
    # bernoulli_ joins the pinned-Philox op set: native CPU replay draws
    # from the pinned counters and the shard path reproduces any row
    # range of a full materialization bitwise.
    from torch.nn import Module, Parameter

    from torchdistx_amd import _C, deferred_init

    class M(Module):
        def __init__(self):
            super().__init__()
            w = torch.empty(96, 32)
            w.bernoulli_(0.3)
            self.p = Parameter(w)

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(303)
        full_m = deferred_init(M)
        full = _C.materialize_tensor(full_m.p).detach()
        vals = full.unique().tolist()
        assert set(vals) <= {0.0, 1.0}
        frac = full.mean().item()
        assert 0.25 < frac < 0.35  # p = 0.3 over 3072 samples

        torch.manual_seed(303)
        part_m = deferred_init(M)
        for start, end in [(0, 96), (7, 23), (95, 96)]:
            shard = _C.materialize_tensor_shard(part_m.p, start, end)
            assert torch.equal(shard, full[start:end]), (start, end)
    finally:
        _C.set_native_init_cpu(False)


def test_bernoulli_stock_replay_matches_eager() -> None:
    # Without the native path, bernoulli_ replays through the stock
    # generator cursor and stays bitwise-equal to eager construction.
    from torch.nn import Module, Parameter

    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.empty(32, 8).bernoulli_(0.5))

    torch.manual_seed(404)
    m = deferred_init(M)
    materialize_module(m)
    torch.manual_seed(404)
    e = M()
    assert torch.equal(m.p, e.p)


def test_broadcast_preserves_tied_parameters_multiproc() -> None:
    # Tied parameters must land as ONE tensor object per rank after a
    # bucketed broadcast (receivers included).
    from tests._dist_utils import run_distributed

    run_distributed(_tied_broadcast_worker, world_size=2)


def _tied_broadcast_worker(rank, world_size):
    import torch
    from torch.nn import Linear, Parameter

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_module_distributed

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            p = Parameter(torch.empty(8, 8).normal_())
            self.a = Linear(8, 8, bias=False)
            self.b = Linear(8, 8, bias=False)
            self.a.weight = p
            self.b.weight = p

    torch.manual_seed(900 + rank)  # rank-skewed: only the wire reconciles
    m = deferred_init(M)
    assert m.a.weight is m.b.weight
    _C.set_native_init_cpu(True)
    try:
        materialize_module_distributed(m, mode="broadcast")
    finally:
        _C.set_native_init_cpu(False)
    assert m.a.weight is m.b.weight, "tied param split by broadcast"


def test_allgather_and_shard_tied_params_multiproc() -> None:
    from tests._dist_utils import run_distributed

    run_distributed(_tied_shard_worker, world_size=2)


def _tied_shard_worker(rank, world_size):
    import torch
    from torch.nn import Linear, Parameter

    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_module_distributed

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            p = Parameter(torch.empty(8, 8).normal_())
            self.a = Linear(8, 8, bias=False)
            self.b = Linear(8, 8, bias=False)
            self.a.weight = p
            self.b.weight = p

    torch.manual_seed(901)
    m = deferred_init(M)
    _C.set_native_init_cpu(True)
    try:
        owner_map = materialize_module_distributed(m, mode="shard")
    finally:
        _C.set_native_init_cpu(False)
    # One unique tensor -> one ownership entry; on the owner both slots
    # hold the same object, elsewhere both stay fake.
    assert len(owner_map) == 1
    owner = owner_map[0]
    if rank == owner:
        assert m.a.weight is m.b.weight
        assert not _C.can_materialize(m.a.weight)
    else:
        assert _C.can_materialize(m.a.weight)


def test_broadcast_bucket_boundaries_multiproc() -> None:
    # Many tensors, two dtypes, and a tiny bucket budget: forces multiple
    # buckets per (owner, dtype) and exercises the pack/unpack offsets
    # across bucket splits at world 2.
    from tests._dist_utils import run_distributed

    results = run_distributed(_bucket_boundary_worker, world_size=2)
    assert results[0] == results[1]  # digests agree across ranks


def _bucket_boundary_worker(rank, world_size):
    import hashlib

    import torch
    from torch.nn import Module, Parameter

    import torchdistx_amd.parallel.sharded_materialize as sm
    from torchdistx_amd import _C, deferred_init
    from torchdistx_amd.parallel import materialize_module_distributed

    class M(Module):
        def __init__(self):
            super().__init__()
            for i in range(13):
                dt = torch.float32 if i % 2 else torch.bfloat16
                n = [3, 17, 64, 129, 1000][i % 5]
                self.register_parameter(
                    f"p{i}", Parameter(torch.empty(n, dtype=dt).normal_())
                )

    saved = sm._BUCKET_BYTES
    sm._BUCKET_BYTES = 512  # force many small buckets
    try:
        torch.manual_seed(777 + rank)  # skewed: the wire must reconcile
        m = deferred_init(M)
        _C.set_native_init_cpu(True)
        try:
            materialize_module_distributed(m, mode="broadcast")
        finally:
            _C.set_native_init_cpu(False)
    finally:
        sm._BUCKET_BYTES = saved

    h = hashlib.sha256()
    for name, p in sorted(m.named_parameters()):
        h.update(name.encode())
        h.update(p.detach().view(torch.uint8).numpy().tobytes())
    return h.hexdigest()


def test_broadcast_pipeline_fuzz() -> None:
    # Bounded slice of scripts/broadcast_fuzz.py: random models (counts,
    # shapes, dtypes, tied params), random bucket budgets, rank-skewed
    # seeds half the time, worlds 2-4 — every rank must converge to one
    # digest through the bucketed wire.
    import os
    import subprocess
    import sys as _sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    if not os.path.exists(os.path.join(repo, "scripts", "broadcast_fuzz.py")):
        pytest.skip("scripts/ not present (wheel-installed run)")
    result = subprocess.run(
        [_sys.executable, "scripts/broadcast_fuzz.py", "6", "100"],
        capture_output=True, text=True, timeout=500,
        cwd=repo,
    )
    assert result.returncode == 0, result.stdout[-1500:] + result.stderr[-500:]
