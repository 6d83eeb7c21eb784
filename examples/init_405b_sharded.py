#!/usr/bin/env python3
"""Initialize a model larger than any single GPU: Llama-3-405B (812 GB
bf16) across one MI355X node via slice materialization — each rank
materializes only its contiguous dim-0 row-slice of every parameter
(~101 GB), bitwise-consistent with the full model, zero communication.

  torchrun --standalone --nproc-per-node 8 examples/init_405b_sharded.py

Set SMALL=1 to run the same flow on Llama-3-8B (e.g. on one GPU).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch
import torch.distributed as dist

from torchdistx_amd import deferred_init
from torchdistx_amd.models import LLAMA3_8B, LLAMA3_405B, build_model
from torchdistx_amd.parallel import materialize_module_dim0_sharded
from torchdistx_amd.utils import describe_module


def main():
    if "RANK" not in os.environ:  # plain `python ...` -> single-rank run
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("LOCAL_RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29572")
    dist.init_process_group("nccl")
    rank = dist.get_rank()
    world = dist.get_world_size()
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))

    cfg = LLAMA3_8B if os.environ.get("SMALL") else LLAMA3_405B

    # Fail fast with a clear message instead of an allocator OOM when the
    # per-rank slice cannot fit (e.g. the full 812 GB model at world 1).
    per_rank = -(-cfg.n_params * 2 // world)  # bf16 bytes, ceil
    hbm = torch.cuda.get_device_properties(0).total_memory
    if per_rank > hbm:
        if rank == 0:
            print(f"FATAL: {cfg.name} needs {per_rank / 1e9:.0f} GB per "
                  f"rank at world {world} but the device has "
                  f"{hbm / 1e9:.0f} GB; run with more ranks, or SMALL=1 "
                  f"for the Llama-3-8B variant.", file=sys.stderr)
        dist.destroy_process_group()
        sys.exit(1)

    torch.manual_seed(0)  # all ranks pin the same Philox streams
    t0 = time.perf_counter()
    model = deferred_init(build_model, cfg, device="cuda",
                          dtype=torch.bfloat16)
    t1 = time.perf_counter()
    if rank == 0:
        d = describe_module(model)
        print(f"recorded {d['n_recorded_tensors']} tensors, "
              f"{d['pending_bytes'] / 1e9:.0f} GB pending, "
              f"record took {(t1 - t0) * 1e3:.0f} ms")

    shards = materialize_module_dim0_sharded(model, rank, world)
    torch.cuda.synchronize()
    t2 = time.perf_counter()

    local_gb = sum(s.numel() * s.element_size() for s in shards.values()) / 1e9
    print(f"rank {rank}: {local_gb:.1f} GB materialized in "
          f"{(t2 - t1) * 1e3:.0f} ms")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
