"""bench.py driver contract: one JSON line with the required fields, CPU
plumbing config (BASELINE config 1)."""

import json
import subprocess
import sys

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
