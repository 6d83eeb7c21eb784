# Randomized differential test of the deferred-init tape: generate random
# op programs (factories, in-place mutation, views, aliasing, copies), run
# them once eagerly and once recorded under deferred_init, materialize the
# surviving tensors in random order, and require exact equality. This
# covers the storage-alias-aware replay logic (the reference's hardest
# code, deferred_init.cc:506-667) far beyond the hand-written cases.

import random

import pytest
import torch
from torch.nn import Module

from torchdistx_amd.deferred_init import deferred_init, materialize_tensor


def _build_program(rng: random.Random, n_steps: int):
    """Returns a list of (op_name, args) executed by _run_program."""
    ops = []
    n_tensors = 0

    def any_tensor():
        return rng.randrange(n_tensors)

    for _ in range(n_steps):
        if n_tensors == 0 or rng.random() < 0.25:
            ops.append(("new", rng.choice([4, 6, 8]), rng.uniform(-2, 2)))
            n_tensors += 1
            continue
        r = rng.random()
        if r < 0.25:
            ops.append(("view", any_tensor()))
            n_tensors += 1
        elif r < 0.4:
            ops.append(("slice", any_tensor(), rng.randrange(2)))
            n_tensors += 1
        elif r < 0.6:
            ops.append(("add_", any_tensor(), rng.uniform(-1, 1)))
        elif r < 0.7:
            ops.append(("mul_", any_tensor(), rng.uniform(0.5, 1.5)))
        elif r < 0.8:
            ops.append(("fill_", any_tensor(), rng.uniform(-3, 3)))
        elif r < 0.9:
            ops.append(("copy_", any_tensor(), any_tensor()))
        else:
            ops.append(("addt", any_tensor(), any_tensor()))
            n_tensors += 1
    return ops


def _run_program(ops):
    ts = []
    for op in ops:
        kind = op[0]
        if kind == "new":
            _, n, val = op
            ts.append(torch.full([n, n], val))
        elif kind == "view":
            ts.append(ts[op[1]].view(-1))
        elif kind == "slice":
            base = ts[op[1]]
            half = base.shape[0] // 2
            ts.append(base[half:] if op[2] else base[:half])
        elif kind == "add_":
            ts[op[1]].add_(op[2])
        elif kind == "mul_":
            ts[op[1]].mul_(op[2])
        elif kind == "fill_":
            ts[op[1]].fill_(op[2])
        elif kind == "copy_":
            dst, src = ts[op[1]], ts[op[2]]
            if dst.numel() == src.numel():
                dst.copy_(src.reshape(dst.shape))
        elif kind == "addt":
            a, b = ts[op[1]], ts[op[2]]
            if a.numel() == b.numel():
                ts.append(a + b.reshape(a.shape))
            else:
                ts.append(a + 1)
    return ts


@pytest.mark.parametrize("seed", range(30))
def test_random_program_replay_matches_eager(seed) -> None:
    rng = random.Random(seed)
    ops = _build_program(rng, n_steps=25)

    eager = _run_program(ops)

    class Holder(Module):
        def __init__(self):
            super().__init__()
            self.tensors = _run_program(ops)

    holder = deferred_init(Holder)

    order = list(range(len(eager)))
    rng.shuffle(order)
    for i in order:
        got = materialize_tensor(holder.tensors[i])
        assert torch.equal(got, eager[i]), (seed, i, ops)
