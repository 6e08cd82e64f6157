"""Native C++ event-loop engine (csrc/engine_native.cpp) on MI355X:
P=1 equivalence against the sequential reference, multi-worker async
semantics, SAGA history, and the straggler model."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.native import NativeLocalEngine
from asyncframework_amd.engine.worker import Shard
from asyncframework_amd.ops import torch_ref
from asyncframework_amd.utils.philox import bernoulli_mask


def _cfg(**kw):
    base = dict(d=64, N=20_000, num_workers=1, num_iterations=60, gamma=0.3,
                taw=1 << 30, batch_rate=0.05, bucket_ratio=0.5,
                printer_freq=1 << 30, delay_coeff=0.0, seed=42,
                device="cuda:0", snapshot_weights=False)
    base.update(kw)
    return EngineConfig(**base)


def _shards(cfg, X, y):
    out = []
    for s, t in row_shards(cfg.N, cfg.num_workers):
        out.append(Shard(row_start=s, n_rows=t - s, X=X[s:t], y=y[s:t]))
    return out


def test_native_p1_matches_sequential_ref():
    cfg = _cfg()
    X, y = synthetic_dense(cfg.N, cfg.d, seed=1, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run()
    assert res["k"] == cfg.num_iterations
    w = torch.zeros(cfg.d, device="cuda:0")
    for k in range(cfg.num_iterations):
        mask = torch.from_numpy(
            bernoulli_mask(cfg.seed, k + 1, 0, cfg.N, cfg.batch_rate)).cuda()
        g, _ = torch_ref.grad_dense(X.float(), y, w, mask, cfg.objective)
        gamma_k = cfg.gamma / math.sqrt(k // cfg.num_workers + 1)
        w -= gamma_k * g / cfg.par_recs
    rel = float((eng.w - w).norm() / (w.norm() + 1e-12))
    assert rel < 1e-4, rel


def test_native_multiworker_async():
    cfg = _cfg(num_workers=4, num_iterations=400, bucket_ratio=0.5,
               N=40_000)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations
    assert res["applied"] >= cfg.num_iterations
    obj0 = float(((X.float() @ torch.zeros(cfg.d, device="cuda:0") - y) ** 2).mean())
    obj1 = float(((X.float() @ eng.w - y) ** 2).mean())
    assert obj1 < obj0


def test_native_tau_rejects():
    cfg = _cfg(num_workers=4, num_iterations=150, taw=0, bucket_ratio=0.3,
               N=40_000)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=3, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations


def test_native_saga():
    cfg = _cfg(algo="asaga", gamma=0.05, num_workers=2, num_iterations=200,
               N=40_000)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=4, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations
    obj0 = float(((X.float() @ torch.zeros(cfg.d, device="cuda:0") - y) ** 2).mean())
    obj1 = float(((X.float() @ eng.w - y) ** 2).mean())
    assert obj1 < obj0
    assert any(int((a != 0).sum()) > 0 for a in eng.alpha_tables)


def test_native_delay_model():
    cfg = _cfg(num_workers=8, num_iterations=300, delay_coeff=-1.0,
               calib_factor=3, N=80_000)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=5, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations
    assert res["avg_delay_ms"] > 0  # calibration activated


def test_native_bench_marks():
    cfg = _cfg(num_workers=4, N=40_000, num_iterations=10 ** 9)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=6, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    elapsed, res = eng.bench(warmup=50, steps=200, max_wall_s=120)
    assert elapsed > 0
    assert res["k"] >= 250


def test_native_cli_driver_contract(capsys):
    """asgd-thread --engine native: full stdout contract through the C++
    event loop (snapshots, waiting times, objective sweep)."""
    from asyncframework_amd.cli import drivers
    drivers.asgd_thread(["synthetic", "synthetic", "64", "20000", "4", "60",
                         "0.3", "1000000", "0.05", "0.5", "20", "0", "42",
                         "--device", "cuda:0", "--engine", "native"])
    out = capsys.readouterr().out
    lines = out.splitlines()
    assert lines[-1] == "finished"
    assert any(l.startswith("Iteration ") for l in lines)
    assert "Individual waiting times:" in lines
    import re
    sep = max(i for i, l in enumerate(lines) if l.startswith("*********"))
    csv = [l for l in lines[sep + 1:] if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    assert len(csv) >= 2
    objs = [float(l.split(",")[1]) for l in csv]
    assert objs[-1] < objs[0]


def test_native_host_spill_matches_hbm():
    """Spill mode (BASELINE config 5 on the native engine): pinned-host
    master + mask-keyed staging refresh must produce the same run as the
    HBM-resident history at P=1 (deterministic ordering)."""
    runs = {}
    for placement in ("device", "host"):
        cfg = _cfg(algo="asaga", gamma=0.05, num_iterations=80,
                   history_placement=placement)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=3, device="cuda:0")
        eng = NativeLocalEngine(cfg, _shards(cfg, X, y),
                                torch.device("cuda:0"))
        res = eng.run(max_wall_s=120)
        assert res["k"] == cfg.num_iterations
        torch.cuda.synchronize()
        runs[placement] = (eng.w.clone(), eng.alpha_tables[0].cpu().clone())
    w_dev, a_dev = runs["device"]
    w_host, a_host = runs["host"]
    assert eng.alpha_tables[0].is_pinned()  # the master really is host DRAM
    rel = float((w_dev - w_host).norm() / (w_dev.norm() + 1e-12))
    assert rel < 1e-5, rel
    assert int((a_host != 0).sum()) > 0
    assert torch.allclose(a_dev.cpu(), a_host, atol=1e-6)


def test_native_host_spill_multiworker():
    cfg = _cfg(algo="asaga", gamma=0.05, num_workers=4, num_iterations=200,
               N=40_000, history_placement="host")
    X, y = synthetic_dense(cfg.N, cfg.d, seed=4, device="cuda:0")
    eng = NativeLocalEngine(cfg, _shards(cfg, X, y), torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations
    assert all(t.is_pinned() for t in eng.alpha_tables)
    assert any(int((t != 0).sum()) > 0 for t in eng.alpha_tables)


def test_native_csr_saga_wave_matches_nowave_p1():
    """CSR SAGA at P=1 is a deterministic sequential iteration: the wave
    dispatch path (grad_csr_wave_kernel + saga_commit_wave_kernel) must
    reproduce the per-worker singleton path (same kernels, same Philox
    keys; only atomic accumulation order may differ)."""
    import os
    from asyncframework_amd.data.synthetic import synthetic_csr

    def run(no_wave):
        if no_wave:
            os.environ["ASYNCAMD_NO_WAVE"] = "1"
        else:
            os.environ.pop("ASYNCAMD_NO_WAVE", None)
        cfg = _cfg(algo="asaga", gamma=0.05, num_workers=1,
                   num_iterations=120, N=20_000, d=512, batch_rate=0.02)
        indptr, indices, values, y = synthetic_csr(
            cfg.N, cfg.d, seed=11, device="cuda:0")
        sh = Shard(row_start=0, n_rows=cfg.N, indptr=indptr,
                   indices=indices, values=values, y=y)
        eng = NativeLocalEngine(cfg, [sh], torch.device("cuda:0"))
        res = eng.run(max_wall_s=120)
        assert res["k"] == cfg.num_iterations
        return eng.w.clone(), eng.alpha_tables[0].clone()

    try:
        w_wave, a_wave = run(no_wave=False)
        w_ref, a_ref = run(no_wave=True)
    finally:
        os.environ.pop("ASYNCAMD_NO_WAVE", None)
    # identical sampled-row sets => identical alpha nonzero patterns
    assert torch.equal(a_wave != 0, a_ref != 0)
    arel = float((a_wave - a_ref).norm() / (a_ref.norm() + 1e-12))
    assert arel < 2e-3, arel
    rel = float((w_wave - w_ref).norm() / (w_ref.norm() + 1e-12))
    assert rel < 1e-3, rel


def test_native_dense_wave_matches_nowave_p1():
    """Dense ASGD at P=1 is deterministic and sequential: the wave path
    (grad_dense_wave_kernel) must reproduce the singleton path bit-for-bit
    up to atomic accumulation order."""
    import os

    def run(no_wave):
        if no_wave:
            os.environ["ASYNCAMD_NO_WAVE"] = "1"
        else:
            os.environ.pop("ASYNCAMD_NO_WAVE", None)
        cfg = _cfg(num_iterations=150, N=30_000, d=256, batch_rate=0.02,
                   gamma=0.2)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=13, device="cuda:0")
        eng = NativeLocalEngine(cfg, _shards(cfg, X, y),
                                torch.device("cuda:0"))
        res = eng.run(max_wall_s=120)
        assert res["k"] == cfg.num_iterations
        return eng.w.clone()

    try:
        w_wave = run(no_wave=False)
        w_ref = run(no_wave=True)
    finally:
        os.environ.pop("ASYNCAMD_NO_WAVE", None)
    rel = float((w_wave - w_ref).norm() / (w_ref.norm() + 1e-12))
    assert rel < 1e-4, rel


def test_native_csr_saga_with_delays():
    """CSR SAGA + cloud straggler model: delayed workers release through
    the SINGLETON dispatch path while others go through waves — exercises
    the 1-slot commit branch (dispatch_impl) mixing with wave commits."""
    from asyncframework_amd.data.synthetic import synthetic_csr
    cfg = _cfg(algo="asaga", gamma=0.05, num_workers=8, num_iterations=400,
               N=80_000, d=256, batch_rate=0.02, delay_coeff=-1.0,
               calib_factor=3)
    indptr, indices, values, y = synthetic_csr(cfg.N, cfg.d, seed=17,
                                               device="cuda:0")
    shards = []
    for s, t in row_shards(cfg.N, cfg.num_workers):
        base = int(indptr[s])
        shards.append(Shard(
            row_start=s, n_rows=t - s,
            indptr=(indptr[s:t + 1] - base).contiguous(),
            indices=indices[base:int(indptr[t])],
            values=values[base:int(indptr[t])], y=y[s:t]))
    eng = NativeLocalEngine(cfg, shards, torch.device("cuda:0"))
    res = eng.run(max_wall_s=120)
    assert res["k"] >= cfg.num_iterations
    assert res["avg_delay_ms"] > 0  # calibration activated => delays ran
    assert any(int((a != 0).sum()) > 0 for a in eng.alpha_tables)
