"""Multiproc fuzz of the bucketed-broadcast pipeline over gloo: random
models (tensor counts/shapes/dtypes, tied params), random bucket budgets
and world sizes; every rank must converge to identical digests, and with
uniform seeds to the local native materialization. Usage:
python scripts/broadcast_fuzz.py [n_seeds] [seed_base]"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

from tests._dist_utils import run_distributed  # noqa: E402


def _worker(rank, world_size, seed):
    import hashlib
    import random

    import torch
    from torch.nn import Module, Parameter

    import torchdistx_amd.parallel.sharded_materialize as sm
    from torchdistx_amd import _C, deferred_init

    rng = random.Random(seed)
    n_tensors = rng.randint(1, 20)
    specs = []
    for i in range(n_tensors):
        shape = rng.choice([(3,), (17,), (64,), (33, 9), (128, 5), (7, 3, 5)])
        dtype = rng.choice([torch.float32, torch.bfloat16, torch.float16])
        kind = rng.choice(["normal", "uniform", "zeros", "tied"])
        specs.append((shape, dtype, kind))

    class M(Module):
        def __init__(self):
            super().__init__()
            prev = None
            for i, (shape, dtype, kind) in enumerate(specs):
                if kind == "tied" and prev is not None:
                    setattr(self, f"p{i}", prev)
                    continue
                t = torch.empty(shape, dtype=dtype)
                if kind == "uniform":
                    t.uniform_(-1, 1)
                elif kind == "zeros":
                    t.zero_()
                else:
                    t.normal_()
                p = Parameter(t)
                setattr(self, f"p{i}", p)
                prev = p

    skew = rng.random() < 0.5  # half the seeds: only the wire reconciles
    torch.manual_seed(seed + (rank * 7919 if skew else 0))
    m = deferred_init(M)

    saved = sm._BUCKET_BYTES
    sm._BUCKET_BYTES = rng.choice([256, 4096, 1 << 20])
    _C.set_native_init_cpu(True)
    try:
        sm.materialize_module_distributed(m, mode="broadcast")
    finally:
        _C.set_native_init_cpu(False)
        sm._BUCKET_BYTES = saved

    h = hashlib.sha256()
    ids = {}
    for name, p in sorted(m.named_parameters()):
        h.update(name.encode())
        h.update(p.detach().view(torch.uint8).numpy().tobytes())
        ids.setdefault(id(p), []).append(name)
    n_tied_groups = sum(1 for v in ids.values() if len(v) > 1)
    return h.hexdigest(), n_tied_groups, skew


def main():
    n_seeds = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    base = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    import random

    fails = 0
    for seed in range(base, base + n_seeds):
        world = random.Random(seed).choice([2, 3, 4])
        results = run_distributed(_worker, world, seed)
        digests = {r[0] for r in results}
        if len(digests) != 1:
            fails += 1
            print(f"FAIL seed {seed} world {world}: rank digests diverge")
        if (seed - base + 1) % 10 == 0:
            print(f"{seed - base + 1}/{n_seeds}, {fails} fails", flush=True)
    print(f"done: {fails} fails over {n_seeds} broadcast-fuzz seeds")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
