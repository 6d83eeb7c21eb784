"""Extended soak of the slice-materialization fuzzer (see
tests/test_slice_fuzz.py). Usage:
python scripts/slice_soak.py [n_seeds] [device] [seed_base]"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

from test_slice_fuzz import _run_case
from torchdistx_amd import _C

n_seeds = int(sys.argv[1]) if len(sys.argv) > 1 else 1000
device = sys.argv[2] if len(sys.argv) > 2 else "cpu"
seed_base = int(sys.argv[3]) if len(sys.argv) > 3 else 0

if device == "cpu":
    _C.set_native_init_cpu(True)

fails = 0
for seed in range(seed_base, seed_base + n_seeds):
    try:
        _run_case(seed, device)
    except AssertionError as e:
        fails += 1
        print("FAIL", seed, e)
        if fails > 5:
            break
    if (seed - seed_base) % 1000 == 999:
        print(f"{seed + 1 - seed_base}/{n_seeds} seeds, {fails} fails",
              flush=True)

print(f"done: {fails} fails over {n_seeds} slice seeds on {device}")
sys.exit(1 if fails else 0)
