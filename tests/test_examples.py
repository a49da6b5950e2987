"""Every shipped example must stay runnable (the reference keeps its
examples executable in CI via stripped notebooks; ours are plain scripts,
so run them)."""

import pathlib
import subprocess
import sys

import pytest

pytestmark = [pytest.mark.torch, pytest.mark.slow]

EXAMPLES = sorted((pathlib.Path(__file__).parent.parent / "examples").glob("*.py"))


@pytest.mark.parametrize("script", EXAMPLES, ids=lambda p: p.name)
def test_example_runs(script):
    proc = subprocess.run(
        [sys.executable, str(script)],
        capture_output=True,
        text=True,
        timeout=600,
    )
    assert proc.returncode == 0, f"{script.name} failed:\n{proc.stderr[-2000:]}"
