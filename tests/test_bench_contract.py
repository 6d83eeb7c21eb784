# Guards the driver contract of bench.py: one JSON line on stdout with the
# agreed fields, runnable without flags on a GPU-less machine.

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(*extra):
    result = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--steps", "2",
         "--warmup", "1", *extra],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
    )
    assert result.returncode == 0, result.stderr
    lines = [l for l in result.stdout.strip().splitlines() if l.strip()]
    assert len(lines) == 1, f"expected exactly one JSON line: {lines}"
    return json.loads(lines[0])


def test_bench_json_contract() -> None:
    d = _run()
    for key in (
        "metric", "value", "unit", "n_gpus", "steps", "warmup",
        "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
        "dtype", "data", "config",
    ):
        assert key in d, key
    assert d["n_gpus"] == 1
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] in ("weak", "strong")
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["model"] == "tiny"
    assert "synthetic" in d["data"]


def test_bench_modes_report_scaling() -> None:
    assert _run("--mode", "replicate")["scaling"] == "weak"
    assert _run("--mode", "slice")["scaling"] == "strong"


@pytest.mark.gpu
def test_bench_json_contract_gpu() -> None:
    # Same contract, exercised on the GPU path (native kernels required,
    # device-memory guard, cuda synchronize bracketing).
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs a ROCm GPU")
    d = _run()
    assert d["config"]["device"] == "cuda"
    assert d["config"]["native_init_kernels"] is True
    assert d["value"] > 0


def test_bench_selftest_world1() -> None:
    result = subprocess.run(
        [sys.executable, "bench.py", "--selftest"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert result.returncode == 0, result.stderr
    d = json.loads(result.stdout.strip().splitlines()[-1])
    assert d["selftest"] == "pass", d


def test_bench_selftest_world2_torchrun() -> None:
    # Rehearses the exact launcher + rank plumbing the driver uses for
    # multi-GPU benches, over gloo on CPU: communicator creation, every
    # materialization mode's cross-rank bitwise contract, shard coverage
    # and slice reassembly at world 2.
    result = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29779", "bench.py", "--selftest"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    line = [l for l in result.stdout.splitlines() if '"selftest"' in l][-1]
    d = json.loads(line)
    assert d["selftest"] == "pass", d
    names = {c["name"] for c in d["checks"]}
    assert "broadcast-skewed-cross-rank" in names
    assert "shard-coverage" in names
