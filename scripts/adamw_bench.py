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
