"""Extended differential fuzz soak of the deferred-init tape (the CI test
runs 30 seeds; this runs thousands). Usage: python scripts/fuzz_soak.py
[n_seeds] [device] [seed_base]."""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import torch
from torch.nn import Module

from test_tape_fuzz import _build_program, _run_program
from torchdistx_amd.deferred_init import deferred_init, materialize_tensor

n_seeds = int(sys.argv[1]) if len(sys.argv) > 1 else 1000
device = sys.argv[2] if len(sys.argv) > 2 else "cpu"
seed_base = int(sys.argv[3]) if len(sys.argv) > 3 else 0
threaded = len(sys.argv) > 4 and sys.argv[4] == "threaded"
if threaded:
    from test_tape_fuzz import _check_seed_threaded

    fails = 0
    for seed in range(seed_base, seed_base + n_seeds):
        try:
            _check_seed_threaded(seed, device)
        except Exception as e:
            fails += 1
            print("FAIL", seed, repr(e)[:300])
        if (seed - seed_base + 1) % 500 == 0:
            print(f"{seed - seed_base + 1}/{n_seeds} threaded seeds, {fails} fails", flush=True)
    print(f"done: {fails} fails over {n_seeds} threaded seeds on {device}")
    sys.exit(1 if fails else 0)

fails = 0
for seed in range(seed_base, seed_base + n_seeds):
    rng = random.Random(seed * 7919 + 13)
    ops = _build_program(rng, 45)
    eager = _run_program(ops, device)

    class Holder(Module):
        def __init__(self):
            super().__init__()
            self.tensors = _run_program(ops, device)

    holder = deferred_init(Holder)
    order = list(range(len(eager)))
    rng.shuffle(order)
    for i in order:
        got = materialize_tensor(holder.tensors[i])
        if not torch.equal(got, eager[i]):
            fails += 1
            print("FAIL", seed, i)
            break
    if seed % 500 == 499:
        print(f"{seed + 1 - seed_base}/{n_seeds} seeds, {fails} fails")

print(f"done: {fails} fails over {n_seeds} seeds on {device}")
sys.exit(1 if fails else 0)
