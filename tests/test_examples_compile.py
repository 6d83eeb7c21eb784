# The examples only *run* on a GPU box (they use the RCCL backend), so on
# CPU CI we at least guarantee they byte-compile — a syntax/import-level
# regression would otherwise surface only on GPU hardware.

import os
import py_compile

import pytest

_EXAMPLES = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples"
)


@pytest.mark.parametrize(
    "name",
    [
        "train_fsdp_slowmo.py",
        "train_fsdp2_anyprecision.py",
        "init_405b_sharded.py",
        "init_from_checkpoint.py",
    ],
)
def test_example_compiles(name: str) -> None:
    py_compile.compile(os.path.join(_EXAMPLES, name), doraise=True)
