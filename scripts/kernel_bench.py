"""Measures every tdx:: kernel's sustained rate on one MI355X — the
reproducible source of docs/performance.md's kernel table. Runs a warm
loop per kernel (DVFS ramps over the first ~15-20 ms of load on a fresh
lease: profiles/dvfs_ramp_note.md).

Usage: python scripts/kernel_bench.py [GiB]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torchdistx_amd  # noqa: F401  (registers tdx:: ops)

assert torch.cuda.is_available()
GIB = float(sys.argv[1]) if len(sys.argv) > 1 else 4.0
NBYTES = int(GIB * (1 << 30))


def rate(fn, bytes_per_iter, iters=10, warm=4):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return bytes_per_iter * iters / (time.perf_counter() - t0) / 1e12


def main():
    rows = []
    bf = torch.empty(NBYTES // 2, dtype=torch.bfloat16, device="cuda")
    f32 = torch.empty(NBYTES // 4, dtype=torch.float32, device="cuda")
    f16 = torch.empty(NBYTES // 2, dtype=torch.float16, device="cuda")

    # store-only ceiling
    rows.append(("hipMemset ceiling", rate(lambda: bf.zero_(), NBYTES)))
    rows.append(("tdx::fill_ bf16",
                 rate(lambda: torch.ops.tdx.fill_(bf, 1.5), NBYTES)))
    for name, t in (("bf16", bf), ("f32", f32), ("f16", f16)):
        rows.append((f"tdx::uniform_ {name}",
                     rate(lambda t=t: torch.ops.tdx.uniform_(
                         t, 0., 1., seed=1, offset=4), NBYTES)))
        rows.append((f"tdx::normal_ {name}",
                     rate(lambda t=t: torch.ops.tdx.normal_(
                         t, 0., 1., seed=1, offset=4), NBYTES)))
    rows.append(("tdx::bernoulli_ bf16",
                 rate(lambda: torch.ops.tdx.bernoulli_(
                     bf, 0.5, seed=1, offset=4), NBYTES)))
    # cast copies (read + write traffic)
    src = torch.empty(NBYTES // 4, dtype=torch.bfloat16, device="cuda")
    dst = torch.empty(NBYTES // 4, dtype=torch.float32, device="cuda")
    torch.ops.tdx.uniform_(src, 0., 1., seed=1, offset=4)
    rows.append(("tdx::copy_ bf16->f32 (rw)",
                 rate(lambda: torch.ops.tdx.copy_(dst, src, False),
                      src.numel() * 6)))
    rows.append(("tdx::copy_ f32->bf16 (rw)",
                 rate(lambda: torch.ops.tdx.copy_(src, dst, False),
                      src.numel() * 6)))

    for name, tbps in rows:
        print(f"{name:<28} {tbps:6.2f} TB/s")


if __name__ == "__main__":
    main()
