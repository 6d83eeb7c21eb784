#!/usr/bin/env python3
"""Flagship benchmark: deferred_init -> materialize wall-clock (the
BASELINE.json metric) on the named model configs.

One step = one full pipeline: record the module-construction tape under
deferred_init (CPU-side, C++ tape), then materialize every parameter and
buffer into HBM3E through the CDNA4 init kernels.

Default config: Llama-3-70B bf16, mode=replicate — every rank materializes
a full replica from the shared Philox streams with ZERO communication
(weak scaling: per-GPU work is fixed as N grows). On the xGMI fabric local
regeneration beats any broadcast by ~40x (see
torchdistx_amd/parallel/sharded_materialize.py), so this mode is both the
fastest way to replicate a model across a node and the honest flagship
number. --mode shard / broadcast measure the FSDP-style sharded path and
the RCCL-broadcast path.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
"""

import argparse
import json
import os
import resource
import sys
import time

import torch


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", type=str, default=None,
                   help="config name (default: llama3-70b on GPU, tiny on CPU)")
    p.add_argument("--mode", type=str, default="replicate",
                   choices=["replicate", "shard", "broadcast", "allgather", "slice"])
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32", "fp16"])
    p.add_argument("--init", type=str, default="fast",
                   choices=["fast", "stock"],
                   help="fast: one normal_ per weight; stock: PyTorch's "
                        "default kaiming resets (uniform_-dominated tape)")
    return p.parse_args()


def main():
    args = get_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        device = "cpu"

    distributed = world > 1
    if distributed:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl" if use_cuda else "gloo")
    else:
        dist = None

    # Fail loudly if materialization would silently fall back to stock ATen
    # kernels on a GPU box.
    if use_cuda:
        os.environ.setdefault("TDX_REQUIRE_NATIVE_INIT", "1")

    from torchdistx_amd import deferred_init, is_deferred, _kernels
    from torchdistx_amd.models import CONFIGS, build_model
    from torchdistx_amd.parallel import (
        materialize_module_dim0_sharded,
        materialize_module_distributed,
    )

    if use_cuda and not _kernels.available():
        print("FATAL: torchdistx_amd._K (CDNA4 kernels) not loaded on a GPU box",
              file=sys.stderr)
        sys.exit(1)

    model_name = args.model or ("llama3-70b" if use_cuda else "tiny")
    cfg = CONFIGS[model_name]
    if args.init != cfg.init:
        import dataclasses

        cfg = dataclasses.replace(cfg, init=args.init)
    dtype = {"bf16": torch.bfloat16, "fp32": torch.float32,
             "fp16": torch.float16}[args.dtype]

    if use_cuda:
        # Fail fast with a clear message instead of an allocator OOM deep
        # inside materialization (e.g. llama3-405b needs >1 rank for any
        # non-replicated mode; replicate of it fits nowhere).
        per_rank = cfg.n_params * dtype.itemsize
        if args.mode in ("shard", "slice"):
            per_rank = -(-per_rank // world)  # this rank's share (ceil)
        hbm = torch.cuda.get_device_properties(0).total_memory
        if per_rank > hbm:  # Mixtral's 281 GB of 288 GB is fine; 812 GB is not
            print(
                f"FATAL: {model_name} mode={args.mode} needs "
                f"{per_rank / 1e9:.0f} GB per rank but the device has "
                f"{hbm / 1e9:.0f} GB; use more ranks or mode=slice/shard",
                file=sys.stderr,
            )
            sys.exit(1)

    def one_step(seed: int) -> None:
        torch.manual_seed(seed)
        module = deferred_init(build_model, cfg, device=device, dtype=dtype)
        if args.mode == "slice":
            # sub-tensor FSDP/TP-style init: this rank materializes its
            # contiguous dim-0 row-slice of every parameter.
            shards = materialize_module_dim0_sharded(module, rank, world)
            assert shards
            del shards
        else:
            materialize_module_distributed(module, mode=args.mode)
            if args.mode != "shard":
                assert not is_deferred(module), "materialization incomplete"
        # Reference-counting frees the module and its tape immediately (the
        # tape's ownership graph is acyclic by design); no gc.collect().
        del module

    def barrier_sync():
        if distributed:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        one_step(seed=1000 + i)
    barrier_sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(seed=2000 + i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks.
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000.0
    n_params = cfg.n_params
    if args.mode == "replicate":
        # Whole-job aggregate: every rank materializes a full replica.
        total_params_per_step = n_params * world
        scaling = "weak"
    else:
        # One model materialized per step across all ranks.
        total_params_per_step = n_params
        scaling = "strong"
    value = total_params_per_step / (elapsed / args.steps)

    rss_gb = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1e6

    if rank == 0:
        print(json.dumps({
            "metric": "deferred_init_materialize_params_per_s",
            "value": value,
            "unit": "params/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": scaling,
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic (random-init weights, no checkpoints)",
            "config": {
                "model": cfg.name,
                "n_params": n_params,
                "mode": args.mode,
                "init": args.init,
                "device": "cuda" if use_cuda else "cpu",
                "native_init_kernels": _kernels.available(),
                "peak_host_rss_gb": round(rss_gb, 2),
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"{args.mode}{world}",
            },
        }))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
