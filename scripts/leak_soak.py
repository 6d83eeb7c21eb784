"""Stability soak: repeated deferred_init -> materialize cycles of one
model, tracking host RSS and device memory. The tape's ownership graph is
acyclic by design, so dropping the module must free everything without
gc; any drift here is a leak. Usage:
python scripts/leak_soak.py [n_iters] [model]"""
import os
import resource
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchdistx_amd import deferred_init, materialize_module
from torchdistx_amd.models import CONFIGS, build_model

n_iters = int(sys.argv[1]) if len(sys.argv) > 1 else 50
model = sys.argv[2] if len(sys.argv) > 2 else "llama3-8b"
device = "cuda" if torch.cuda.is_available() else "cpu"
cfg = CONFIGS[model]


def rss_gb():
    return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1e6


def dev_gb():
    if device != "cuda":
        return 0.0, 0.0
    return (torch.cuda.memory_allocated() / 1e9,
            torch.cuda.memory_reserved() / 1e9)


samples = []
for i in range(n_iters):
    torch.manual_seed(i)
    m = deferred_init(build_model, cfg, device=device,
                      dtype=torch.bfloat16 if device == "cuda"
                      else torch.float32)
    materialize_module(m)
    del m
    if device == "cuda":
        torch.cuda.synchronize()
    if i % 10 == 0 or i == n_iters - 1:
        alloc, reserved = dev_gb()
        samples.append((i, rss_gb(), alloc, reserved))
        print(f"iter {i:4d}: peak_rss {rss_gb():6.2f} GB  "
              f"dev_alloc {alloc:6.2f} GB  dev_reserved {reserved:6.2f} GB",
              flush=True)

# After the final del, allocated device memory must return to ~zero and
# peak RSS must not have kept growing past early-iteration steady state.
final_alloc, _ = dev_gb()
assert final_alloc < 1.0, f"device memory leak: {final_alloc:.2f} GB live"
early_peak = samples[1][1] if len(samples) > 1 else samples[0][1]
assert samples[-1][1] < early_peak * 1.15, (
    f"host RSS drift: {early_peak:.2f} -> {samples[-1][1]:.2f} GB"
)
print(f"leak soak OK: {n_iters} cycles of {model} on {device}, "
      f"final dev_alloc {final_alloc:.3f} GB, peak RSS {samples[-1][1]:.2f} GB")
