"""Leak soak over the round-2 materialization paths: batched planner,
thread-parallel, and world-1 bucketed broadcast cycles, asserting
allocated device memory returns to baseline after each cycle.
Usage: python scripts/leak_soak_paths.py [n_cycles] [model]"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from torchdistx_amd import (
    deferred_init,
    materialize_module,
    materialize_module_batched,
    materialize_module_parallel,
)
from torchdistx_amd.models import CONFIGS, build_model
from torchdistx_amd.parallel import materialize_module_distributed

n_cycles = int(sys.argv[1]) if len(sys.argv) > 1 else 30
model = sys.argv[2] if len(sys.argv) > 2 else "llama3-8b"
assert torch.cuda.is_available()
cfg = CONFIGS[model]

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29790")
dist.init_process_group("nccl", rank=0, world_size=1)

def _tp_windowed(m):
    # Row-parallel windowed slices of every 2-d weight (rank 1 of 4):
    # the result dict goes out of scope, so device memory must return
    # to baseline like the module paths.
    from torchdistx_amd.parallel import materialize_tensor_shard

    for _, p in m.named_parameters():
        if p.dim() == 2:
            n = p.shape[1]
            materialize_tensor_shard(p, n // 4, n // 2, dim=1)


PATHS = {
    "sequential": materialize_module,
    "batched": materialize_module_batched,
    "parallel": lambda m: materialize_module_parallel(m, num_threads=4),
    "broadcast": lambda m: materialize_module_distributed(
        m, mode="broadcast"
    ),
    "tp_windowed": _tp_windowed,
}

baseline = None
worst = 0.0
for i in range(n_cycles):
    for name, fn in PATHS.items():
        torch.manual_seed(i)
        m = deferred_init(build_model, cfg, device="cuda",
                          dtype=torch.bfloat16)
        fn(m)
        torch.cuda.synchronize()
        del m
        alloc = torch.cuda.memory_allocated()
        if baseline is None:
            baseline = alloc
        worst = max(worst, alloc - baseline)
    if (i + 1) % 10 == 0:
        print(f"{i+1}/{n_cycles} cycles, baseline {baseline/1e6:.1f} MB, "
              f"worst drift {worst/1e6:.1f} MB", flush=True)

dist.destroy_process_group()
limit = 64 << 20  # stream pools, comm buffers etc. may retain a little
print(f"done: worst drift {worst/1e6:.1f} MB over {n_cycles} cycles "
      f"x {len(PATHS)} paths")
sys.exit(0 if worst < limit else 1)
