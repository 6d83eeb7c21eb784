#!/bin/bash
# The CPU CI pipeline, runnable on any ROCm-PyTorch box with no network:
# .github/workflows/ci.yaml invokes exactly this script, so the CI
# definition is something that has actually been executed, not an
# aspiration. Stages:
#   1. in-tree build (gfx950 cross-compile) + full CPU suite
#   2. ROCm-tagged wheel build, install into a clean venv (system torch),
#      and the suite re-run against the INSTALLED package from a staging
#      dir so the repo checkout cannot shadow it
set -euo pipefail
cd "$(dirname "$0")/.."
REPO="$PWD"

echo "== stage 1: in-tree build + CPU suite =="
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
python -m pytest tests -q -m "not gpu"

echo "== stage 2: wheel + clean-target install =="
rm -rf dist
python setup.py -q bdist_wheel
WHEEL="$(ls "$REPO"/dist/*.whl)"
TARGET="$(mktemp -d)/install"
python -m pip install -q --no-index --no-deps --target "$TARGET" "$WHEEL"

# Run the suite from a staging dir so the repo checkout cannot shadow the
# installed package; PYTHONPATH points at the wheel install alone.
STAGE="$(mktemp -d)"
cp -r tests bench.py pyproject.toml examples "$STAGE/"
rm -rf "$STAGE"/tests/__pycache__
cd "$STAGE"
PYTHONPATH="$TARGET" python - <<EOF
import os, torchdistx_amd
path = os.path.dirname(torchdistx_amd.__file__)
assert path.startswith("$TARGET"), f"resolved outside the wheel install: {path}"
print("imported from", path)
EOF
PYTHONPATH="$TARGET" python -m pytest tests -q -m "not gpu" -p no:cacheprovider
cd "$REPO"
rm -rf "$TARGET" "$STAGE"
echo "== ci_local: all stages green =="
