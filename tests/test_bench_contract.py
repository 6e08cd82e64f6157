"""bench.py driver contract: one JSON line with the required fields, CPU
plumbing config (BASELINE config 1)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def _run(args):
    out = subprocess.run([sys.executable, "bench.py"] + args,
                         capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-1500:]
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_sync_tiny_cpu_contract():
    j = _run(["--model", "sync-tiny-cpu", "--steps", "25", "--warmup", "5"])
    for k in REQUIRED:
        assert k in j, k
    assert j["metric"] == "gradient updates/sec (whole node)"
    assert j["value"] > 0
    assert j["steps"] == 25 and j["warmup"] == 5
    assert j["data"] == "synthetic"
    assert j["config"]["model"] == "sync-tiny-cpu"


def test_async_cpu_contract():
    j = _run(["--model", "asgd-mnist8m", "--device", "cpu", "--engine",
              "threads", "--rows", "4000", "--cols", "64", "--dtype",
              "fp32", "--steps", "60", "--warmup", "10"])
    assert j["value"] > 0
    assert j["higher_is_better"] is True
    assert j["config"]["parallelism"].startswith("async-ps")


@pytest.mark.timeout(300)
@pytest.mark.parametrize("dist_engine", ["native", "python"])
def test_bench_torchrun_dist_cpu_contract(dist_engine):
    """bench.py's N>1 path end-to-end under torch.distributed.run (2 ranks,
    gloo on CPU): the exact launch topology the driver uses for the scale
    runs, including the default C++ dist server control plane."""
    import subprocess
    args = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node=2", "--master-addr=127.0.0.1",
            "--master-port=29741" if dist_engine == "native"
            else "--master-port=29742",
            "bench.py", "--device", "cpu", "--rows", "4000", "--cols", "32",
            "--steps", "150", "--warmup", "30", "--workers", "4",
            "--dist-engine", dist_engine]
    out = subprocess.run(args, capture_output=True, text=True, timeout=240,
                         cwd=ROOT)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    j = json.loads(line)
    assert j["n_gpus"] == 2
    assert j["value"] > 0
    assert j["steps"] == 150


def test_to_target_summary_dense_cpu():
    """The BASELINE metric's second clause (wall-clock-to-target-loss) is
    computed from the optVars ring: verify target detection, the
    always-defined 95%-of-achieved-progress clause, and the None guard."""
    import torch

    import bench

    class A:
        sparse = False
        objective = "lsq"

    torch.manual_seed(0)
    X = torch.randn(500, 16)
    w_true = torch.randn(16)
    y = X @ w_true
    # opt_vars: w converging toward w_true -> objective decreasing
    opt_vars = [(0, torch.zeros(16))] + [
        (100 * (i + 1), w_true * (i + 1) / 5.0) for i in range(5)]
    out = bench.to_target_summary(A(), (X, y), opt_vars, target_frac=0.5)
    assert out is not None
    assert out["obj_initial"] > out["obj_final"]
    assert out["ms_to_target"] is not None  # exact solve crosses 0.5*obj0
    assert out["ms_to_95pct_of_achieved_progress"] is not None
    assert out["ms_to_95pct_of_achieved_progress"] <= 500
    # guard: too-short rings yield None, not a crash
    assert bench.to_target_summary(A(), (X, y), [(0, torch.zeros(16))]) is None
    assert bench.to_target_summary(A(), (X, y), None) is None
