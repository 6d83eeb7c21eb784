"""A/B of the AnyPrecisionAdamW step on GPU: eager op-sequence vs fused
CDNA4 kernel, on the flagship dtype layout (bf16 params/grad/variance/comp,
fp32 momentum, Kahan on)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from torchdistx_amd.optimizers import AnyPrecisionAdamW

assert torch.cuda.is_available()
n = 1 << 28  # 268M params (~0.5 GB bf16)


def bench(use_fused, steps=10):
    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.randn(n, device="cuda", dtype=torch.bfloat16))
    opt = AnyPrecisionAdamW(
        [p], lr=1e-3, weight_decay=0.01, use_kahan_summation=True,
        momentum_dtype=torch.float32, variance_dtype=torch.bfloat16,
        compensation_buffer_dtype=torch.bfloat16, use_fused=use_fused,
    )
    p.grad = torch.randn_like(p)
    opt.step()  # init state + warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        opt.step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    # bytes: p r/w (2+2) g r (2) m r/w (4+4) v r/w (2+2) c r/w (2+2) = 22 B/elem
    tbps = 22.0 * n / dt / 1e12
    return dt * 1e3, tbps


for name, fused in (("eager", False), ("fused", True)):
    ms, tb = bench(fused)
    print(f"{name}: {ms:8.2f} ms/step  effective {tb:5.2f} TB/s")


def bench_many_small(use_batched, n_tensors=512, numel=16384, steps=10):
    """Launch-overhead regime: hundreds of small params per step."""
    torch.manual_seed(1)
    params = [
        torch.nn.Parameter(
            torch.randn(numel, device="cuda", dtype=torch.bfloat16)
        )
        for _ in range(n_tensors)
    ]
    import torchdistx_amd.optimizers.anyprecision_optimizer as apo

    saved = apo._BATCH_MAX_NUMEL
    apo._BATCH_MAX_NUMEL = (1 << 20) if use_batched else 0
    try:
        opt = AnyPrecisionAdamW(
            params, lr=1e-3, weight_decay=0.01, use_kahan_summation=True,
            momentum_dtype=torch.float32, variance_dtype=torch.bfloat16,
            compensation_buffer_dtype=torch.bfloat16,
        )
        for p in params:
            p.grad = torch.randn_like(p)
        opt.step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            opt.step()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / steps * 1e3
    finally:
        apo._BATCH_MAX_NUMEL = saved


for name, batched in (("per-tensor fused", False), ("batched", True)):
    ms = bench_many_small(batched)
    print(f"512x16k params, {name}: {ms:8.2f} ms/step")
