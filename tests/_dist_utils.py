# Helpers for multi-process (gloo) distributed tests that run on CPU-only
# machines. Mirrors the role of the reference's FSDPTest harness
# (reference tests/python/test_comm_hooks_fsdp.py:18-24) without requiring
# GPUs: node topology is simulated by declaring small subgroups as "nodes"
# (reference test_comm_hooks_fsdp.py:476-487).

import os
import pickle
import tempfile

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        result = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(result, f)
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world_size, *args, port=None):
    """Runs ``fn(rank, world_size, *args)`` in ``world_size`` processes over
    gloo and returns the list of per-rank return values (must be
    picklable)."""
    if port is None:
        import socket

        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
    with tempfile.TemporaryDirectory() as result_dir:
        mp.spawn(
            _worker,
            args=(world_size, port, fn, args, result_dir),
            nprocs=world_size,
            join=True,
        )
        results = []
        for rank in range(world_size):
            with open(os.path.join(result_dir, f"rank{rank}.pkl"), "rb") as f:
                results.append(pickle.load(f))
        return results
