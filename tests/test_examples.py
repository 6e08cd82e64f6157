"""The runnable examples must stay runnable (they are documentation)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("script", ["verb_layer_driver.py", "engine_api.py"])
def test_example_runs(script):
    out = subprocess.run([sys.executable,
                          os.path.join(ROOT, "examples", script)],
                         capture_output=True, text=True, timeout=240,
                         cwd=ROOT)
    assert out.returncode == 0, out.stderr[-1500:]
    assert out.stdout.strip()
