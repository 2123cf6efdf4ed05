"""The examples/ scripts must run end to end (CPU-sized here; the same
scripts pick GPU shapes when CUDA is present)."""

import pathlib
import subprocess
import sys

import pytest

EX = sorted((pathlib.Path(__file__).parent.parent / "examples").glob("*.py"))


@pytest.mark.parametrize("script", EX, ids=[p.name for p in EX])
@pytest.mark.timeout(300)
def test_example_runs(script):
    r = subprocess.run([sys.executable, str(script)], capture_output=True,
                       text=True, timeout=280,
                       cwd=str(script.parent.parent))
    assert r.returncode == 0, r.stderr[-1500:]
