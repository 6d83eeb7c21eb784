"""Measures the windowed shard kernels (any-dim slice materialization)
on one MI355X — the TP-init hot path. A dim-1 slice of a [R, C] weight
is R contiguous global ranges (one per row), so the windowed kernel's
group-indexed fast path must sustain near the flat kernel's rate; the
odd-width geometry exercises the elementwise fallback.

Usage: python scripts/win_kernel_bench.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torchdistx_amd  # noqa: F401  (registers tdx:: ops)

assert torch.cuda.is_available()

R, C = 28672, 8192  # Llama-3-70B FFN weight shape


def rate(fn, nbytes, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return nbytes * iters / (time.perf_counter() - t0) / 1e12


def main():
    rows = []
    # flat reference: the same bytes via the dim-0 shard kernel
    flat = torch.empty(R // 8, C, dtype=torch.bfloat16, device="cuda")
    n0 = (R // 8) * C
    rows.append(("normal_shard_ dim0 1/8 (flat ref)",
                 rate(lambda: torch.ops.tdx.normal_shard_(
                     flat, 0, n0, 0., 1., seed=1, offset=4), n0 * 2)))
    # dim-1 1/8 slice: aligned windowed fast path (block_len 1024)
    cols = C // 8
    sh = torch.empty(R, cols, dtype=torch.bfloat16, device="cuda")
    rows.append(("normal_shard_win_ dim1 1/8 aligned",
                 rate(lambda: torch.ops.tdx.normal_shard_win_(
                     sh, R, cols, C, 2 * cols, 0., 1., seed=1, offset=4),
                     sh.numel() * 2)))
    rows.append(("uniform_shard_win_ dim1 1/8 aligned",
                 rate(lambda: torch.ops.tdx.uniform_shard_win_(
                     sh, R, cols, C, 2 * cols, 0., 1., seed=1, offset=4),
                     sh.numel() * 2)))
    # odd width: elementwise fallback
    sh2 = torch.empty(R, 1000, dtype=torch.bfloat16, device="cuda")
    rows.append(("normal_shard_win_ odd-width fallback",
                 rate(lambda: torch.ops.tdx.normal_shard_win_(
                     sh2, R, 1000, C, 2048, 0., 1., seed=1, offset=4),
                     sh2.numel() * 2)))
    # f32 aligned (4-elem groups)
    sh3 = torch.empty(R, cols, dtype=torch.float32, device="cuda")
    rows.append(("normal_shard_win_ dim1 1/8 f32",
                 rate(lambda: torch.ops.tdx.normal_shard_win_(
                     sh3, R, cols, C, 2 * cols, 0., 1., seed=1, offset=4),
                     sh3.numel() * 4)))

    for name, tbps in rows:
        print(f"{name:<38} {tbps:6.2f} TB/s")


if __name__ == "__main__":
    main()
