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


@pytest.mark.timeout(300)
def test_compare_async_sync_harness():
    """The async-vs-sync time-to-target-loss harness (the reference's
    headline experiment) stays runnable and emits its JSON contract."""
    import json
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools",
                                      "compare_async_sync.py"),
         "--device", "cpu", "--rows", "1200", "--cols", "24",
         "--workers", "4", "--iters-async", "100", "--gamma-async", "0.3",
         "--rate", "0.2", "--target-rel", "0.5"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-1500:]
    j = json.loads(out.stdout.strip().splitlines()[-1])
    assert j["experiment"] == "async_vs_sync_time_to_target_loss"
    for k in ("initial_objective", "target_objective", "async", "sync",
              "speedup_to_target"):
        assert k in j
    assert j["async"]["time_to_target_ms"] >= 0
    assert j["speedup_to_target"] > 0


@pytest.mark.timeout(300)
def test_plot_loss_workflow(tmp_path):
    """The reference's figure workflow: driver log -> error-vs-time plot."""
    log = tmp_path / "run.log"
    out = subprocess.run(
        [sys.executable, "-m", "asyncframework_amd.cli", "asgd-thread",
         "synthetic", "synthetic", "16", "200", "2", "40", "0.5",
         "1000000", "0.3", "0.5", "10", "0", "42"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0
    log.write_text(out.stdout)
    png = tmp_path / "fig.png"
    out2 = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "plot_loss.py"),
         str(log), "-o", str(png)],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out2.returncode == 0, out2.stderr[-800:]
    assert png.exists() and png.stat().st_size > 5000
