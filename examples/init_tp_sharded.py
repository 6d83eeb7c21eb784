#!/usr/bin/env python3
"""Megatron-style tensor-parallel initialization: each rank materializes
only ITS slice of every TP-sharded weight — column-parallel attention /
up-projections along dim 0, row-parallel output / down-projections along
dim 1 — with zero communication, bitwise-consistent with the full model.

  torchrun --standalone --nproc-per-node 8 examples/init_tp_sharded.py

Also runs single-rank: `python examples/init_tp_sharded.py` (the whole
model is then "this rank's slice").
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch
import torch.distributed as dist

from torchdistx_amd import deferred_init
from torchdistx_amd.models import LLAMA3_8B, TINY, build_model
from torchdistx_amd.parallel import materialize_module_tp_sharded


def tp_shard_dims(model) -> dict:
    """The Megatron split for our transformer blocks: wq/wk/wv and the
    FFN up/gate projections are column-parallel (shard rows, dim 0 of
    the [out, in] weight); wo and the FFN down projection are
    row-parallel (shard columns, dim 1). Embeddings shard on the vocab
    dim; norms and the row-parallel path's biases replicate."""
    dims = {}
    for name, _ in model.named_parameters():
        if name.endswith(("wq.weight", "wk.weight", "wv.weight",
                          "w1.weight", "w3.weight")):
            dims[name] = 0
        elif name.endswith(("wo.weight", "w2.weight")):
            dims[name] = 1
        elif name in ("tok_emb.weight", "lm_head.weight"):
            dims[name] = 0
    return dims


def main():
    if "RANK" not in os.environ:  # plain `python ...` -> single-rank run
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("LOCAL_RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29573")
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cuda":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))

    # Llama-3-8B on GPU; the tiny config keeps a GPU-less rehearsal quick.
    cfg = LLAMA3_8B if device == "cuda" else TINY
    torch.manual_seed(0)  # all ranks pin the same Philox streams
    t0 = time.perf_counter()
    model = deferred_init(build_model, cfg, device=device,
                          dtype=torch.bfloat16)
    dims = tp_shard_dims(model)
    shards = materialize_module_tp_sharded(model, dims, rank, world)
    if device == "cuda":
        torch.cuda.synchronize()
    t1 = time.perf_counter()

    local_gb = sum(s.numel() * s.element_size() for s in shards.values()) / 1e9
    n_sharded = sum(1 for n in shards if n in dims)
    n_repl = len(shards) - n_sharded
    print(f"rank {rank}/{world}: {n_sharded} TP-sharded + {n_repl} "
          f"replicated tensors, {local_gb:.2f} GB local, "
          f"{(t1 - t0) * 1e3:.0f} ms")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
