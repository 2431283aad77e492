"""Every example script runs end-to-end on the CPU path (living docs,
reference examples/ analog)."""

import os
import pathlib
import subprocess
import sys

import pytest

ROOT = pathlib.Path(__file__).resolve().parent.parent
SCRIPTS = sorted(
    str(p.relative_to(ROOT)) for p in (ROOT / "examples").rglob("*.py")
)


@pytest.mark.parametrize("script", SCRIPTS)
@pytest.mark.timeout(300)
def test_example_runs(script):
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [sys.executable, str(ROOT / script)], env=env, timeout=280,
        capture_output=True, text=True,
    )
    assert out.returncode == 0, out.stderr[-1500:]
