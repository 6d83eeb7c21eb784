# Drop-in proof: the REFERENCE repository's own unit tests (read from its
# read-only mount, never copied) pass unmodified against this framework
# through the `torchdistx` alias package.

import os
import subprocess
import sys

import pytest

_REF = "/root/reference/tests/python"


@pytest.mark.skipif(
    not os.path.isdir(_REF), reason="reference mount not present"
)
def test_reference_unit_tests_pass_against_alias() -> None:
    result = subprocess.run(
        [
            sys.executable,
            "-m",
            "pytest",
            os.path.join(_REF, "test_fake.py"),
            os.path.join(_REF, "test_deferred_init.py"),
            "-q",
            "-p",
            "no:cacheprovider",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert result.returncode == 0, result.stdout + result.stderr
    assert "8 passed" in result.stdout
