"""run.py high-level API: dataset routing, worker construction, and the
EngineConfig derived knobs (reference SparkASGDThread.scala:39-51, 188,
233-237; MLUtils.loadLibSVMFile routing in the run templates)."""

import numpy as np
import pytest
import torch

from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.run import (build_csr_workers, build_dense_workers,
                                    load_dataset, run_engine)


def test_gate_floor_behavior():
    # floor(P * beta): the reference's quorum (SparkASGDThread.scala:233-237)
    assert EngineConfig(num_workers=8, bucket_ratio=0.93).gate == 7
    assert EngineConfig(num_workers=8, bucket_ratio=1.0).gate == 8
    assert EngineConfig(num_workers=8, bucket_ratio=0.0).gate == 0
    assert EngineConfig(num_workers=3, bucket_ratio=0.5).gate == 1
    assert EngineConfig(num_workers=1, bucket_ratio=0.93).gate == 0


def test_par_recs():
    cfg = EngineConfig(d=10, N=1000, num_workers=4, batch_rate=0.2)
    assert cfg.par_recs == pytest.approx(50.0)


def test_torch_dtype_mapping():
    assert EngineConfig(dtype="fp32").torch_dtype() is torch.float32
    assert EngineConfig(dtype="bf16").torch_dtype() is torch.bfloat16
    assert EngineConfig(dtype="fp16").torch_dtype() is torch.float16
    assert EngineConfig(dtype="fp64").torch_dtype() is torch.float64
    with pytest.raises(KeyError):
        EngineConfig(dtype="int8").torch_dtype()


def test_load_dataset_synthetic_dense_and_csr():
    cfg = EngineConfig(d=12, N=64, seed=3, objective="lsq", dtype="fp32")
    X, y = load_dataset(cfg, "synthetic", "synthetic")
    assert X.shape == (64, 12) and y.shape == (64,)
    indptr, indices, values, ys = load_dataset(cfg, "synthetic", "synthetic",
                                               sparse=True)
    assert indptr.shape == (65,) and ys.shape == (64,)
    assert indices.shape == values.shape


def test_load_dataset_libsvm_routing(tmp_path):
    p = tmp_path / "toy.libsvm"
    p.write_text("1 1:0.5 3:2.0\n-1 2:1.5\n1 1:1.0 4:4.0\n-1 3:0.25\n")
    cfg = EngineConfig(d=4, N=4, dtype="fp32")
    X, y = load_dataset(cfg, str(tmp_path) + "/", "toy.libsvm")
    assert X.shape == (4, 4)
    # LibSVM is 1-indexed: "1:0.5" -> column 0
    assert float(X[0, 0]) == 0.5 and float(X[1, 1]) == 1.5
    indptr, indices, values, ys = load_dataset(
        cfg, str(tmp_path) + "/", "toy.libsvm", sparse=True)
    dense = torch.zeros(4, 4)
    for r in range(4):
        for j in range(int(indptr[r]), int(indptr[r + 1])):
            dense[r, int(indices[j])] += values[j]
    assert torch.equal(dense, X)
    assert torch.equal(ys, y)


def test_build_dense_workers_cover_all_rows():
    cfg = EngineConfig(d=8, N=37, num_workers=3)
    X = torch.randn(37, 8)
    y = torch.randn(37)
    workers = build_dense_workers(cfg, X, y)
    assert len(workers) == 3
    total = sum(w.shard.n_rows for w in workers)
    assert total == 37
    # shard boundaries 4-aligned except possibly the first start (0 is)
    for w in workers:
        assert w.shard.row_start % 4 == 0
    got = torch.cat([w.shard.X for w in workers])
    assert torch.equal(got, X)


def test_build_csr_workers_rebase_indptr():
    cfg = EngineConfig(d=6, N=9, num_workers=2)
    # 9 rows, each row r has one entry at col r % 6 with value r
    indptr = torch.arange(10, dtype=torch.int32)
    indices = torch.tensor([r % 6 for r in range(9)], dtype=torch.int32)
    values = torch.arange(9, dtype=torch.float32)
    y = torch.ones(9)
    workers = build_csr_workers(cfg, indptr, indices, values, y)
    assert len(workers) == 2
    for w in workers:
        assert int(w.shard.indptr[0]) == 0
        assert int(w.shard.indptr[-1]) == w.shard.values.shape[0]
    # second shard's values still carry the global row numbers
    s1 = workers[1].shard
    assert float(s1.values[0]) == float(s1.row_start)


def test_run_engine_rejects_native_sync():
    cfg = EngineConfig(d=4, N=16, num_workers=2, num_iterations=4, sync=True)
    X, y = load_dataset(cfg, "synthetic", "synthetic")
    workers = build_dense_workers(cfg, X, y)
    with pytest.raises(AssertionError):
        run_engine(cfg, workers, engine="native")


def test_require_hip_raises_without_extension(monkeypatch):
    """On a GPU box a missing _hip_core must fail LOUDLY, not silently fall
    back to eager torch (driver checks which .so the GPU tests loaded)."""
    from asyncframework_amd import ops
    monkeypatch.setattr(ops, "_load_hip", lambda: None)
    monkeypatch.delenv("ASYNCAMD_ALLOW_FALLBACK", raising=False)
    with pytest.raises(RuntimeError, match="mandatory"):
        ops._require_hip("grad_dense")
    # explicit debug escape hatch
    monkeypatch.setenv("ASYNCAMD_ALLOW_FALLBACK", "1")
    assert ops._require_hip("grad_dense") is None


def test_synthetic_dense_deterministic():
    from asyncframework_amd.data.synthetic import synthetic_dense
    X1, y1 = synthetic_dense(300, 16, seed=5)
    X2, y2 = synthetic_dense(300, 16, seed=5)
    X3, _ = synthetic_dense(300, 16, seed=6)
    assert torch.equal(X1, X2) and torch.equal(y1, y2)
    assert not torch.equal(X1, X3)
    # logistic labels are deterministic {0,1} (no RNG draw for labels,
    # reproducible across devices)
    _, yl = synthetic_dense(300, 16, seed=5, objective="logistic")
    assert set(yl.unique().tolist()) <= {0.0, 1.0}


def test_synthetic_csr_deterministic():
    from asyncframework_amd.data.synthetic import synthetic_csr
    a = synthetic_csr(200, 64, seed=3)
    b = synthetic_csr(200, 64, seed=3)
    for t1, t2 in zip(a, b):
        assert torch.equal(t1, t2)


def test_graph_engine_rejects_unsupported_configs():
    """Unsupported combos fail LOUDLY at construction (no silent
    feature-dropping): host-spill history and straggler injection belong
    to the threads/native engines."""
    from asyncframework_amd.engine.graph import GraphEngine
    from asyncframework_amd.engine.worker import Shard
    sh = Shard(row_start=0, n_rows=8, X=torch.zeros(8, 4),
               y=torch.zeros(8))
    cfg = EngineConfig(d=4, N=8, num_workers=1, algo="asaga",
                       history_placement="host")
    with pytest.raises(AssertionError, match="host-spill"):
        GraphEngine(cfg, sh, torch.device("cpu"))
    cfg2 = EngineConfig(d=4, N=8, num_workers=1, delay_coeff=1.0)
    with pytest.raises(AssertionError, match="straggler"):
        GraphEngine(cfg2, sh, torch.device("cpu"))


def test_hip_adapter_guard_rails():
    """The GPU adapter validates its inputs loudly (CPU tensors, misaligned
    shard starts) — importable and checkable without a GPU since the .so
    cross-compiles."""
    pytest.importorskip("asyncframework_amd._hip_core")
    from asyncframework_amd.ops import hip as hip_ops
    X = torch.zeros(8, 4)
    y = torch.zeros(8)
    w = torch.zeros(4)
    out = torch.zeros(4)
    with pytest.raises(AssertionError, match="CUDA"):
        hip_ops.grad_dense(X, y, w, out, seed=1, round_k=0, row_start=0,
                           rate=0.5, obj=0)
