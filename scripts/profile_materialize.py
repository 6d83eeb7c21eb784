"""Phase breakdown + cProfile of one deferred_init->materialize step on GPU."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cProfile
import pstats
import sys
import time

import torch

from torchdistx_amd import deferred_init
from torchdistx_amd.deferred_init import materialize_module
from torchdistx_amd.models import CONFIGS, build_model

model_name = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
cfg = CONFIGS[model_name]
dev = "cuda" if torch.cuda.is_available() else "cpu"

def step(seed):
    torch.manual_seed(seed)
    t0 = time.perf_counter()
    m = deferred_init(build_model, cfg, device=dev, dtype=torch.bfloat16)
    t1 = time.perf_counter()
    materialize_module(m)
    if dev == "cuda":
        torch.cuda.synchronize()
    t2 = time.perf_counter()
    del m
    import gc
    gc.collect()
    if dev == "cuda":
        torch.cuda.synchronize()
    t3 = time.perf_counter()
    return (t1 - t0, t2 - t1, t3 - t2)

# warmup
step(0)
for i in range(3):
    r, mzt, fr = step(i + 1)
    print(f"step{i}: record={r*1e3:.1f}ms materialize={mzt*1e3:.1f}ms free={fr*1e3:.1f}ms")

pr = cProfile.Profile()
pr.enable()
step(99)
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("cumulative")
st.print_stats(25)
