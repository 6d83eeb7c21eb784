#!/usr/bin/env python3
"""FSDP2 end-to-end on MI355X: deferred_init -> fully_shard_deferred
(unit-by-unit materialization through the CDNA4 init kernels, then
DTensor sharding) -> bf16 training with AnyPrecisionAdamW.

Run on one node (1..8 GPUs):
  torchrun --standalone --nproc-per-node 8 examples/train_fsdp2_anyprecision.py

Set BIG=1 for Llama-3-8B instead of the tiny demo config.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
import torch.nn.functional as F

from torchdistx_amd import deferred_init
from torchdistx_amd.models import LLAMA3_8B, TINY, build_model
from torchdistx_amd.optimizers import AnyPrecisionAdamW
from torchdistx_amd.parallel import fully_shard_deferred


def main():
    if "RANK" not in os.environ:  # plain `python ...` -> single-rank run
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("LOCAL_RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29573")
    dist.init_process_group("nccl")
    rank = dist.get_rank()
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))

    cfg = LLAMA3_8B if os.environ.get("BIG") else TINY

    torch.manual_seed(0)  # same tape -> bitwise-identical init on every rank
    model = deferred_init(build_model, cfg, device="cuda",
                          dtype=torch.bfloat16)
    # Materialize + shard one transformer block at a time: peak memory
    # beyond the final shards is a single block's full parameters.
    fully_shard_deferred(model, submodules=list(model.blocks))

    optim = AnyPrecisionAdamW(
        model.parameters(),
        lr=3e-4,
        momentum_dtype=torch.bfloat16,
        variance_dtype=torch.bfloat16,
        use_kahan_summation=True,
    )

    seq = min(256, cfg.max_seq_len)
    for step in range(10):
        tokens = torch.randint(0, cfg.vocab_size, (2, seq), device="cuda")
        logits = model(tokens[:, :-1])
        loss = F.cross_entropy(
            logits.float().flatten(0, 1), tokens[:, 1:].flatten()
        )
        optim.zero_grad()
        loss.backward()
        optim.step()
        if rank == 0:
            print(f"step {step}: loss {loss.item():.4f}")

    torch.cuda.synchronize()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
