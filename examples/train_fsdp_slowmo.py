#!/usr/bin/env python3
"""End-to-end example: the workflow torchdistx exists for, on MI355X.

deferred_init constructs the model weightless -> FSDP wraps it (its
built-in torchdistx support materializes through the CDNA4 init kernels)
-> the SlowMo communication hook keeps gradient traffic on the intra-node
xGMI fabric -> AnyPrecisionAdamW trains in bf16 with fp32-quality updates
through the fused step kernel.

Run on one node (1..8 GPUs):
  torchrun --standalone --nproc-per-node 8 examples/train_fsdp_slowmo.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
from torch.distributed.fsdp import FullyShardedDataParallel as FSDP
from torch.distributed.fsdp import ShardingStrategy

from torchdistx_amd import deferred_init
from torchdistx_amd.models import LLAMA3_8B, TINY, build_model
from torchdistx_amd.optimizers import AnyPrecisionAdamW
from torchdistx_amd.slowmo import SlowMomentumOptimizer, SlowMoState, slowmo_hook


def main():
    if "RANK" not in os.environ:  # plain `python ...` -> single-rank run
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("LOCAL_RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group("nccl")
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)

    cfg = LLAMA3_8B if os.environ.get("BIG") else TINY

    torch.manual_seed(0)  # same tape -> bitwise-identical init on every rank
    model = deferred_init(build_model, cfg, device="cuda", dtype=torch.bfloat16)

    # FSDP detects the deferred parameters (via the torchdistx alias
    # package) and materializes them through the native kernels.
    fsdp = FSDP(
        model,
        sharding_strategy=ShardingStrategy.NO_SHARD,
        device_id=local_rank,
    )
    state = SlowMoState(subgroup=None, sync_grads=True)  # one subgroup/node
    fsdp.register_comm_hook(state, slowmo_hook)

    optim = SlowMomentumOptimizer(
        base_optim=AnyPrecisionAdamW(
            fsdp.parameters(),
            lr=3e-4,
            weight_decay=0.1,
            use_kahan_summation=True,
            momentum_dtype=torch.float32,
            variance_dtype=torch.bfloat16,
        ),
        slowmo_freq=48,
        slowmo_factor=0.5,
        slowmo_lr=1.0,
    )

    for step in range(10):
        seq = min(256, cfg.max_seq_len)
        tokens = torch.randint(0, cfg.vocab_size, (4, seq), device="cuda")
        optim.zero_grad()
        logits = fsdp(tokens[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(),
            tokens[:, 1:].reshape(-1),
        )
        loss.backward()
        optim.step()
        if dist.get_rank() == 0:
            print(f"step {step}: loss {loss.item():.4f}")

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
