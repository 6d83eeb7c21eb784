# Differential fuzz of SLICE materialization boundary arithmetic: random
# simple init chains (factory -> whole-tensor RNG/fill -> detach
# passthroughs), random ranges along a RANDOM dim, odd lengths — the
# slice-start element offsets deliberately misalign with the kernels'
# 8-element Philox groups, and dim > 0 takes the windowed shard kernels
# (n_blocks > 1). Every slice must be bitwise-equal to the same window of
# a full materialization of an INDEPENDENT tape recorded from the same
# seed.

import random

import pytest
import torch

from torchdistx_amd import deferred_init
from torchdistx_amd import _C

_SHAPES = [
    (1,),
    (7,),
    (8,),
    (9,),
    (64,),
    (1, 1),
    (3, 5),
    (5, 8),
    (8, 3),
    (16, 17),
    (17, 16),
    (2, 3, 5),
    (4, 4, 4),
    (5, 1, 9),
]

_DTYPES = [torch.float32, torch.bfloat16, torch.float16]


def _random_chain(rng, shape, dtype, device):
    kind = rng.choice(
        ["uniform", "normal", "bernoulli", "fill", "zero", "plain"]
    )
    t = torch.empty(shape, dtype=dtype, device=device)
    if kind == "uniform":
        a = rng.uniform(-2.0, 0.0)
        t.uniform_(a, a + rng.uniform(0.1, 3.0))
    elif kind == "normal":
        t.normal_(rng.uniform(-1.0, 1.0), rng.uniform(0.1, 2.0))
    elif kind == "bernoulli":
        t.bernoulli_(rng.uniform(0.0, 1.0))
    elif kind == "fill":
        t.fill_(rng.uniform(-3.0, 3.0))
    elif kind == "zero":
        t.zero_()
    else:  # plain factory via zeros (empty has no defined bits)
        t = torch.zeros(shape, dtype=dtype, device=device)
    # Pointwise-scalar tail ops (trunc_normal_-style chains): slicing
    # must commute with any mix of them.
    for _ in range(rng.randint(0, 3)):
        op = rng.choice(["mul", "add", "clamp", "abs", "erfinv", "neg"])
        if op == "mul":
            t.mul_(rng.uniform(-2.0, 2.0))
        elif op == "add":
            t.add_(rng.uniform(-1.0, 1.0))
        elif op == "clamp":
            lo = rng.uniform(-1.0, 0.0)
            t.clamp_(min=lo, max=lo + rng.uniform(0.1, 2.0))
        elif op == "abs":
            t.abs_()
        elif op == "erfinv":
            # keep the domain in (-1, 1) first
            t.clamp_(min=-0.999, max=0.999)
            t.erfinv_()
        else:
            t.neg_()
    if rng.random() < 0.3:
        t = t.detach()
    return t


def _run_case(seed: int, device: str) -> None:
    rng = random.Random(seed)
    shape = rng.choice(_SHAPES)
    dtype = rng.choice(_DTYPES)

    def build():
        class Holder(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.t = _random_chain(rng2, shape, dtype, device)

        return Holder()

    # Full materialization from one tape...
    rng2 = random.Random(seed + 1)
    torch.manual_seed(seed)
    full = _C.materialize_tensor(deferred_init(build).t)

    # ...random slices along a random dim from fresh, independent tapes
    # (dim > 0 exercises the windowed shard kernels).
    for _ in range(3):
        dim = rng.randrange(len(shape))
        n = shape[dim]
        start = rng.randint(0, n)
        end = rng.randint(start, n)
        rng2 = random.Random(seed + 1)
        torch.manual_seed(seed)
        holder = deferred_init(build)
        shard = _C.materialize_tensor_shard(holder.t, start, end, dim)
        expect_shape = list(shape)
        expect_shape[dim] = end - start
        assert shard.shape == tuple(expect_shape)
        assert torch.equal(shard, full.narrow(dim, start, end - start)), (
            seed, shape, dtype, dim, start, end
        )


@pytest.mark.parametrize("seed", range(60))
def test_slice_fuzz_cpu(seed: int) -> None:
    # CPU reference impls of the shard ops share the Philox layout with
    # the CDNA4 kernels, so the boundary arithmetic is the same code path
    # shape the GPU takes.
    _C.set_native_init_cpu(True)
    try:
        _run_case(seed, "cpu")
    finally:
        _C.set_native_init_cpu(False)


@pytest.mark.gpu
@pytest.mark.parametrize("seed", range(40))
def test_slice_fuzz_gpu(seed: int) -> None:
    if not torch.cuda.is_available():
        pytest.skip("needs a ROCm GPU")
    _run_case(seed, "cuda")
