# Deferred-init tests. Coverage model: reference
# tests/python/test_deferred_init.py (identity/no-op/lifecycle) plus the
# in-place/view replay example from the reference docs
# (fake_tensor_and_deferred_init.rst:197-209), RNG fidelity, .data
# recording, external-tensor version tracking, and materialize_module
# options.

from typing import cast

import pytest
import torch
from torch import Tensor
from torch.nn import Module, Parameter

from torchdistx_amd.deferred_init import (
    deferred_init,
    is_deferred,
    materialize_module,
    materialize_tensor,
)
from torchdistx_amd.fake import fake_mode, is_fake


def test_materialize_tensor_is_noop_for_real_tensors() -> None:
    a = torch.ones([10])
    assert materialize_tensor(a) is a


def test_materialize_tensor_returns_same_tensor() -> None:
    class FooModule(Module):
        def __init__(self):
            super().__init__()
            self.param1 = Parameter(torch.ones([5]))
            self.param2 = self.param1

    module = deferred_init(FooModule)

    a = materialize_tensor(cast(Tensor, module.param1))
    b = materialize_tensor(cast(Tensor, module.param1))
    c = materialize_tensor(cast(Tensor, module.param2))

    assert a is b
    assert a is c
    assert isinstance(a, Parameter)
    assert torch.equal(a.detach(), torch.ones([5]))


def test_is_deferred_returns_right_value() -> None:
    class FooModule(Module):
        def __init__(self):
            super().__init__()
            self.param1 = Parameter(torch.ones([5]))
            self.param2 = Parameter(torch.ones([5]))

    module = FooModule()
    assert not is_deferred(module)

    module = deferred_init(FooModule)
    assert is_deferred(module)

    materialize_module(module)
    assert not is_deferred(module)

    module = deferred_init(FooModule)
    module.param1 = materialize_tensor(module.param1)
    assert is_deferred(module)

    module.param2 = materialize_tensor(module.param2)
    assert not is_deferred(module)


def test_is_deferred_raises_on_wrong_type() -> None:
    with pytest.raises(ValueError):
        is_deferred("not a tensor")  # type: ignore[arg-type]


def test_materialized_module_matches_eager_init() -> None:
    torch.manual_seed(1234)
    deferred = deferred_init(torch.nn.Linear, 16, 32)
    materialize_module(deferred)

    torch.manual_seed(1234)
    eager = torch.nn.Linear(16, 32)

    assert torch.equal(deferred.weight, eager.weight)
    assert torch.equal(deferred.bias, eager.bias)
    assert deferred.weight.requires_grad
    assert isinstance(deferred.weight, Parameter)


def test_materialized_sequential_matches_eager_init() -> None:
    def build():
        return torch.nn.Sequential(
            torch.nn.Linear(8, 8),
            torch.nn.LayerNorm(8),
            torch.nn.Linear(8, 4),
        )

    torch.manual_seed(99)
    deferred = deferred_init(build)
    assert is_deferred(deferred)
    materialize_module(deferred)

    torch.manual_seed(99)
    eager = build()

    for (dn, dp), (en, ep) in zip(
        deferred.named_parameters(), eager.named_parameters()
    ):
        assert dn == en
        assert torch.equal(dp, ep), dn


def test_inplace_view_replay() -> None:
    # The docs' canonical aliasing example: a view must observe in-place
    # updates to its base that happened after the view was created.
    class M(Module):
        def __init__(self):
            super().__init__()
            a = torch.zeros([4])
            v = a.view(2, 2)
            a.add_(1)
            self.p = Parameter(v)

    m = deferred_init(M)
    p = materialize_tensor(cast(Tensor, m.p))
    assert torch.equal(p.detach(), torch.ones([2, 2]))


def test_inplace_after_view_on_view_replay() -> None:
    class M(Module):
        def __init__(self):
            super().__init__()
            a = torch.zeros([4])
            v = a.view(2, 2)
            v.fill_(3)
            self.base = Parameter(a)

    m = deferred_init(M)
    base = materialize_tensor(cast(Tensor, m.base))
    assert torch.equal(base.detach(), torch.full([4], 3.0))


def test_dot_data_assignment_is_replayed() -> None:
    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.empty(3))
            self.p.data.fill_(2.0)
            self.p.data = torch.full([3], 7.0)

    m = deferred_init(M)
    p = materialize_tensor(cast(Tensor, m.p))
    assert torch.equal(p.detach(), torch.full([3], 7.0))


def test_item_is_terminal() -> None:
    class M(Module):
        def __init__(self):
            super().__init__()
            scale = (torch.ones(1) * 0.5).item()
            self.p = Parameter(torch.full([2], scale))

    m = deferred_init(M)
    p = materialize_tensor(cast(Tensor, m.p))
    assert torch.equal(p.detach(), torch.full([2], 0.5))


def test_external_cpu_scalar_version_check() -> None:
    ext = torch.tensor(2.0)

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.zeros([3]) + ext)

    good = deferred_init(M)
    assert torch.equal(
        materialize_tensor(cast(Tensor, good.p)).detach(), torch.full([3], 2.0)
    )

    stale = deferred_init(M)
    ext.add_(1)
    with pytest.raises(RuntimeError, match="modified in place"):
        materialize_tensor(cast(Tensor, stale.p))


def test_materialize_module_buffers_only() -> None:
    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.ones([2]))
            self.register_buffer("b", torch.zeros([2]))

    m = deferred_init(M)
    materialize_module(m, buffers_only=True)
    assert is_fake(m.p)
    assert not is_fake(m.b)
    materialize_module(m)
    assert not is_deferred(m)


def test_materialize_module_check_fn() -> None:
    class Leaf(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.ones([2]))

    class Root(Module):
        def __init__(self):
            super().__init__()
            self.skip = Leaf()
            self.keep = Leaf()

    m = deferred_init(Root)
    materialize_module(m, check_fn=lambda mod: mod is not m.skip)
    assert is_fake(m.skip.p)
    assert not is_fake(m.keep.p)


def test_requires_grad_is_preserved() -> None:
    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.ones([2]), requires_grad=False)
            self.register_buffer("b", torch.zeros([2]))

    m = deferred_init(M)
    materialize_module(m)
    assert not m.p.requires_grad
    assert not m.b.requires_grad


def test_fake_without_record_raises_value_error() -> None:
    with fake_mode():
        a = torch.ones([3])
    with pytest.raises(ValueError):
        materialize_tensor(a)


def test_fake_from_plain_fake_mode_rejected_in_deferred_init() -> None:
    with fake_mode():
        orphan = torch.ones([3])

    def build():
        class M(Module):
            def __init__(self):
                super().__init__()
                self.p = Parameter(orphan + 1)

        return M()

    with pytest.raises(RuntimeError, match="deferred-init"):
        deferred_init(build)


def test_deferred_init_forward_shapes_on_fake() -> None:
    # A forward pass on a fully fake module stays fake and shape-correct.
    def build():
        return torch.nn.Linear(6, 3)

    m = deferred_init(build)
    with fake_mode():
        x = torch.randn(5, 6)
        y = m(x)
    assert is_fake(y)
    assert y.shape == (5, 3)


def test_lazy_modules_with_dry_run() -> None:
    # Reference contract (docs/src/deferred_init.rst:152-170): lazy modules
    # work when the dry-run happens inside the deferred function.
    def my_lazy(out_features):
        m = torch.nn.LazyLinear(out_features)
        m(torch.ones([10, 10]))
        return m

    m = deferred_init(my_lazy, 10)
    assert is_deferred(m)
    assert m.weight.shape == (10, 10)
    materialize_module(m)
    assert m(torch.ones(2, 10)).shape == (2, 10)


def test_inference_tensor_rejected_at_materialization() -> None:
    # Reference contract (docs/src/deferred_init.rst:200-202): inference
    # tensors cannot participate; the error comes at materialization.
    with torch.inference_mode():
        inf = torch.tensor(2.0)

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.zeros(3) + inf)

    m = deferred_init(M)
    with pytest.raises(RuntimeError, match="[Ii]nference"):
        materialize_tensor(m.p)


def test_beyond_hardware_scale_instantiation() -> None:
    # The reference's qualitative capability (fake_tensor.rst:65) at scale:
    # a 405B-parameter model (812 GB bf16) is fully inspectable on this
    # GPU-less CI machine with ~zero memory.
    from torchdistx_amd.models import LLAMA3_405B, build_model

    m = deferred_init(build_model, LLAMA3_405B, device="cpu",
                      dtype=torch.bfloat16)
    assert is_deferred(m)
    n = sum(p.numel() for p in m.parameters())
    assert n == LLAMA3_405B.n_params
    assert n > 400e9
    assert m.blocks[0].attn.wq.weight.shape == (16384, 16384)


def test_external_tensor_copy_into_param() -> None:
    # The checkpoint-loading flow: copying an external (real) tensor into a
    # deferred parameter records and replays faithfully.
    ext = torch.arange(12.0).reshape(3, 4)

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.empty(3, 4))
            self.p.data.copy_(ext)

    m = deferred_init(M)
    p = materialize_tensor(cast(Tensor, m.p))
    assert torch.equal(p.detach(), ext)


def test_fp8_tensors_through_fake_and_deferred() -> None:
    # MI355X is an fp8-first chip (OCP e4m3/e5m2); fp8 tensors flow through
    # the fake and deferred layers (replay uses stock kernels for fp8).
    with fake_mode():
        t = torch.zeros(8, 8, dtype=torch.float8_e4m3fn)
    assert is_fake(t) and t.dtype == torch.float8_e4m3fn

    class M(Module):
        def __init__(self):
            super().__init__()
            self.register_buffer(
                "w8", torch.randn(4, 4).to(torch.float8_e4m3fn)
            )

    m = deferred_init(M)
    w = materialize_tensor(cast(Tensor, m.w8))
    assert w.dtype == torch.float8_e4m3fn
    assert w.float().abs().sum().item() > 0


def test_nested_deferred_init() -> None:
    # A deferred_init inside a deferred_init records into one tape; the
    # nesting is level-counted, not scoped per call.
    class Inner(Module):
        def __init__(self):
            super().__init__()
            self.p = Parameter(torch.full([3], 2.0))

    class Outer(Module):
        def __init__(self):
            super().__init__()
            self.inner = deferred_init(Inner)
            self.q = Parameter(torch.full([3], 5.0))

    m = deferred_init(Outer)
    assert is_deferred(m.inner)
    materialize_module(m)
    assert torch.equal(m.inner.p.detach(), torch.full([3], 2.0))
    assert torch.equal(m.q.detach(), torch.full([3], 5.0))


def test_materialize_during_recording_via_item() -> None:
    # A terminal op mid-recording materializes its inputs while the outer
    # recording continues and stays consistent.
    class M(Module):
        def __init__(self):
            super().__init__()
            a = torch.full([2], 3.0)
            s = a.sum().item()       # forces materialization of `a`'s chain
            self.p = Parameter(a * s)

    m = deferred_init(M)
    p = materialize_tensor(cast(Tensor, m.p))
    assert torch.equal(p.detach(), torch.full([2], 18.0))


def test_saving_deferred_module_raises_loudly() -> None:
    # Fake tensors have no storage; serializing a deferred module fails
    # loudly instead of writing garbage (materialize first).
    import io

    m = deferred_init(torch.nn.Linear, 3, 3)
    with pytest.raises((NotImplementedError, RuntimeError)):
        torch.save(m.state_dict(), io.BytesIO())


# ---------------------------------------------------------------------------
# RNG session semantics (see docs/deferred_init.md "RNG contract").
# ---------------------------------------------------------------------------


def _native_cpu_init(model_fn):
    """Records and materializes through the pinned-Philox CPU-native path."""
    from torchdistx_amd import _C

    m = deferred_init(model_fn)
    _C.set_native_init_cpu(True)
    try:
        materialize_module(m)
    finally:
        _C.set_native_init_cpu(False)
    return m


def test_unseeded_sessions_draw_independent_native_streams() -> None:
    # Two deferred_init sessions without re-seeding must produce different
    # weights on the pinned (native) path, like eager code would.
    torch.manual_seed(1000)
    m1 = _native_cpu_init(lambda: torch.nn.Linear(16, 16))
    m2 = _native_cpu_init(lambda: torch.nn.Linear(16, 16))
    assert not torch.equal(m1.weight, m2.weight)
    assert not torch.equal(m1.bias, m2.bias)


def test_reseeded_sessions_pin_identical_native_streams() -> None:
    # Same seed -> bitwise-identical native init (pure function of the
    # generator state at session entry).
    torch.manual_seed(1001)
    m1 = _native_cpu_init(lambda: torch.nn.Linear(16, 16))
    torch.manual_seed(1001)
    m2 = _native_cpu_init(lambda: torch.nn.Linear(16, 16))
    assert torch.equal(m1.weight, m2.weight)
    assert torch.equal(m1.bias, m2.bias)


def test_unseeded_sessions_differ_on_stock_cpu_replay() -> None:
    # Also holds for the stock-ATen CPU replay path: each session snapshots
    # a different generator state into its replay cursor.
    torch.manual_seed(1002)
    m1 = deferred_init(torch.nn.Linear, 16, 16)
    m2 = deferred_init(torch.nn.Linear, 16, 16)
    materialize_module(m1)
    materialize_module(m2)
    assert not torch.equal(m1.weight, m2.weight)


def test_cpu_replay_does_not_consume_ambient_generator() -> None:
    # Replay draws from the session's snapshot cursor, not the live
    # generator: materialization leaves the ambient RNG stream untouched.
    torch.manual_seed(1003)
    m = deferred_init(torch.nn.Linear, 8, 8)
    before = torch.get_rng_state()
    materialize_module(m)
    after = torch.get_rng_state()
    assert torch.equal(before, after)


def test_concurrent_thread_sessions_are_independent() -> None:
    # Sessions are thread-local: two threads recording at once must not
    # clobber each other's seed/slot state into colliding pins.
    import threading

    torch.manual_seed(1004)
    results = {}
    barrier = threading.Barrier(2)

    def build(key):
        barrier.wait()
        results[key] = _native_cpu_init(lambda: torch.nn.Linear(16, 16))

    threads = [
        threading.Thread(target=build, args=(i,)) for i in range(2)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not torch.equal(results[0].weight, results[1].weight)


def test_use_fused_true_raises_when_unavailable() -> None:
    # use_fused=True must never silently fall back to the eager op
    # sequence (a benchmark would silently measure the wrong path).
    from torchdistx_amd.optimizers import AnyPrecisionAdamW

    p = torch.nn.Parameter(torch.randn(8))
    p.grad = torch.randn(8)
    opt = AnyPrecisionAdamW([p], lr=1e-3, use_fused=True)
    with pytest.raises(RuntimeError, match="use_fused=True"):
        opt.step()


def test_parameter_subclass_is_preserved() -> None:
    # Materialization must restore the ORIGINAL Python class, not degrade
    # custom Parameter subclasses to plain nn.Parameter (mirrors the
    # reference's tp_alloc class preservation).
    class ScaledParameter(Parameter):
        pass

    class M(Module):
        def __init__(self):
            super().__init__()
            self.p = ScaledParameter(torch.ones([4]))

    m = deferred_init(M)
    materialize_module(m)
    assert type(m.p) is ScaledParameter
    assert isinstance(m.p, Parameter)
    assert m.p.requires_grad
    assert torch.equal(m.p.detach(), torch.ones([4]))


def test_tensor_subclass_buffer_is_preserved() -> None:
    class TaggedTensor(Tensor):
        pass

    class M(Module):
        def __init__(self):
            super().__init__()
            self.register_buffer("b", torch.zeros([3]).as_subclass(TaggedTensor))

    m = deferred_init(M)
    materialize_module(m)
    assert type(m.b) is TaggedTensor
    assert torch.equal(torch.Tensor(m.b), torch.zeros([3]))


def test_aliased_subclass_parameters_materialize_to_one_object() -> None:
    class ScaledParameter(Parameter):
        pass

    class M(Module):
        def __init__(self):
            super().__init__()
            p = ScaledParameter(torch.ones([4]))
            self.a = p
            self.b = p

    m = deferred_init(M)
    a = materialize_tensor(cast(Tensor, m.a))
    b = materialize_tensor(cast(Tensor, m.b))
    assert a is b
    assert type(a) is ScaledParameter


def test_materialize_module_parallel_cpu_fallback() -> None:
    # On CPU targets the parallel API takes the sequential tape-order
    # path, preserving the stock-generator eager-parity contract.
    from torchdistx_amd.deferred_init import materialize_module_parallel

    torch.manual_seed(77)
    m = deferred_init(torch.nn.Linear, 8, 8)
    materialize_module_parallel(m, num_threads=4)
    torch.manual_seed(77)
    e = torch.nn.Linear(8, 8)
    assert torch.equal(m.weight, e.weight)
    assert torch.equal(m.bias, e.bias)


def test_checkpoint_load_replaces_deferred_tensors() -> None:
    # The pretrained-weights flow the reference motivates deferred init
    # with: build the module fake (no allocation, no init compute), then
    # load a checkpoint with assign=True — fakes are replaced by the
    # loaded tensors and the recorded init work is simply never run.
    src = torch.nn.Linear(6, 5)
    sd = src.state_dict()

    m = deferred_init(torch.nn.Linear, 6, 5)
    assert is_deferred(m)
    m.load_state_dict(sd, assign=True)
    assert not is_deferred(m)
    assert torch.equal(m.weight, sd["weight"])
    assert torch.equal(m.bias, sd["bias"])
    assert isinstance(m.weight, Parameter) and m.weight.requires_grad

    # Forward runs on the loaded weights.
    x = torch.randn(2, 6)
    assert torch.allclose(m(x), src(x))


def test_partial_checkpoint_load_then_materialize_rest() -> None:
    # Mixed flow: load what the checkpoint has, materialize the rest
    # from the tape. On the pinned-Philox native path the un-loaded
    # parameters get exactly the bits a full materialization would give
    # them — partial replay is order-independent there (on the stock CPU
    # path partial replay draws are ambient-dependent by documented
    # contract, so the native path is the one this flow relies on).
    from torchdistx_amd import _C

    class M(Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(4, 4)
            self.b = torch.nn.Linear(4, 4)

    _C.set_native_init_cpu(True)
    try:
        torch.manual_seed(55)
        full = deferred_init(M)
        materialize_module(full)

        torch.manual_seed(55)
        m = deferred_init(M)
        donor = torch.nn.Linear(4, 4)
        m.a.load_state_dict(donor.state_dict(), assign=True)
        assert not is_deferred(m.a)
        assert is_deferred(m.b)

        materialize_module(m)
        assert not is_deferred(m)
        assert torch.equal(m.a.weight, donor.weight)
        assert torch.equal(m.b.weight, full.b.weight)
        assert torch.equal(m.b.bias, full.b.bias)
    finally:
        _C.set_native_init_cpu(False)


def test_tensor_init_plan_extraction() -> None:
    # Simple chains reduce to their final value step; pointwise tails
    # and cross-tensor dependencies return None (fallback to replay).
    from torchdistx_amd import _C

    class M(Module):
        def __init__(self):
            super().__init__()
            self.n = Parameter(torch.empty(8, 4).normal_(0.5, 2.0))
            self.z = Parameter(torch.zeros(6))
            w = torch.empty(4, 4)
            torch.nn.init.trunc_normal_(w)
            self.t = Parameter(w)  # pointwise tail -> not plannable
            self.d = Parameter(torch.zeros(3) + torch.ones(3))  # dep

    m = deferred_init(M)
    plan = _C.tensor_init_plan(m.n)
    assert plan is not None
    assert plan["kind"] == "normal"
    assert plan["p0"] == 0.5 and plan["p1"] == 2.0
    assert plan["sizes"] == [8, 4]
    assert plan["seed"] != 0
    z = _C.tensor_init_plan(m.z)
    assert z is not None and z["kind"] == "zero"
    assert _C.tensor_init_plan(m.t) is None
    assert _C.tensor_init_plan(m.d) is None
    # planning must not consume the tape:
    materialize_module(m)
    assert not is_deferred(m)


def test_materialize_module_batched_cpu_fallback() -> None:
    from torchdistx_amd.deferred_init import materialize_module_batched

    torch.manual_seed(88)
    m = deferred_init(torch.nn.Linear, 8, 8)
    materialize_module_batched(m)
    torch.manual_seed(88)
    e = torch.nn.Linear(8, 8)
    assert torch.equal(m.weight, e.weight)


def test_rng_on_noncontiguous_view_matches_eager() -> None:
    # The in-place record fast path must compose with views: normal_ on
    # a transposed (non-contiguous) view replays through stock ATen
    # (the native kernels require contiguity and decline) and must
    # reproduce eager bits in tape order.
    class M(Module):
        def __init__(self):
            super().__init__()
            a = torch.zeros(4, 6)
            a.t().normal_(0.0, 1.0)
            self.p = Parameter(a)

    torch.manual_seed(321)
    m = deferred_init(M)
    materialize_module(m)
    torch.manual_seed(321)
    e = M()
    assert torch.equal(m.p, e.p)
