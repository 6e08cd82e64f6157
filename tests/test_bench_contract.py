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
