"""End-to-end engine on one MI355X: async ASGD/ASAGA with the native
kernels, history spill config, and a bench smoke."""

import json
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

from asyncframework_amd import run as runner
from asyncframework_amd.data.synthetic import synthetic_dense, synthetic_csr
from asyncframework_amd.engine.config import EngineConfig


def _cfg(**kw):
    base = dict(d=784, N=50_000, num_workers=2, num_iterations=60,
                gamma=0.5, taw=1 << 30, batch_rate=0.05, bucket_ratio=0.5,
                printer_freq=1 << 30, delay_coeff=0.0, seed=42,
                device="cuda:0", snapshot_weights=False)
    base.update(kw)
    return EngineConfig(**base)


def _obj(X, y, w):
    return float(((X.float() @ w.to(X.device) - y) ** 2).mean())


def test_gpu_async_asgd():
    cfg = _cfg()
    X, y = synthetic_dense(cfg.N, cfg.d, seed=1, dtype=torch.bfloat16,
                           device="cuda:0")
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert res.k >= cfg.num_iterations
    assert _obj(X, y, res.w) < _obj(X, y, torch.zeros(cfg.d, device="cuda:0"))


def test_gpu_async_asaga_hbm_history():
    cfg = _cfg(algo="asaga", gamma=0.05, num_iterations=80)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2, dtype=torch.float32,
                           device="cuda:0")
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert _obj(X, y, res.w) < _obj(X, y, torch.zeros(cfg.d, device="cuda:0"))
    assert any(int((w.alpha != 0).sum()) > 0 for w in workers)


def test_gpu_asaga_host_spill_history():
    """BASELINE config 5: history table in pinned host DRAM, gathered via
    hipMemcpyAsync-backed non_blocking copies each round."""
    cfg = _cfg(algo="asaga", gamma=0.05, num_iterations=40,
               history_placement="host")
    X, y = synthetic_dense(cfg.N, cfg.d, seed=3, dtype=torch.float32,
                           device="cuda:0")
    workers = runner.build_dense_workers(cfg, X, y)
    for w in workers:
        assert w.alpha.device.type == "cpu" and w.alpha.is_pinned()
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert _obj(X, y, res.w) < _obj(X, y, torch.zeros(cfg.d, device="cuda:0"))


def test_gpu_csr_engine():
    cfg = _cfg(d=5000, N=20_000, num_iterations=40, gamma=1.0,
               batch_rate=0.05)
    data = synthetic_csr(cfg.N, cfg.d, nnz_per_row=70, seed=4,
                         device="cuda:0")
    workers = runner.build_csr_workers(cfg, *data)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert res.k >= cfg.num_iterations


def test_gpu_delay_injection_runs():
    """Config 4 shape: straggler injection active (small calib window so the
    flag flips within the run)."""
    cfg = _cfg(delay_coeff=1.0, calib_factor=5, num_iterations=40)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=5, dtype=torch.bfloat16,
                           device="cuda:0")
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert res.k >= cfg.num_iterations


def test_bench_smoke():
    # tight subprocess cap: a wedged engine must fail THIS test in <=2 min,
    # not eat the driver's GPU-tier budget (round-1: 600 s timeout here
    # masked 27 later tests under -x)
    out = subprocess.run(
        [sys.executable, "bench.py", "--rows", "200000", "--steps", "50",
         "--warmup", "10"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    j = json.loads(line)
    assert j["metric"] == "gradient updates/sec (whole node)"
    assert j["value"] > 0
    assert j["n_gpus"] == 1
