"""run.final_report: the shutdown epilogue (elapsed + waiting block +
objective sweep over recorded iterates) for dense AND CSR data — the
reference's SparkASGDThread.scala:346-410 shape."""

import io
import re
from contextlib import redirect_stdout

import torch

from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.local import RunResult
from asyncframework_amd.run import final_report, load_dataset


def _mk_res(cfg, n_snaps=3):
    opt = [(i * 10, torch.randn(cfg.d) * 0.01) for i in range(n_snaps)]
    opt[0] = (0, torch.zeros(cfg.d))
    return RunResult(k=30, elapsed_ms=123, opt_vars=opt,
                     waiting_time={0: 5, 1: 7}, w=opt[-1][1])


def _capture(cfg, res, data, sparse):
    buf = io.StringIO()
    with redirect_stdout(buf):
        final_report(cfg, res, data, sparse)
    return buf.getvalue().splitlines()


def test_final_report_dense():
    cfg = EngineConfig(d=8, N=64, num_workers=2, objective="lsq")
    data = load_dataset(cfg, "synthetic", "synthetic")
    lines = _capture(cfg, _mk_res(cfg), data, sparse=False)
    assert any(re.match(r"Elapsed time\(ms\): 123", l) for l in lines)
    assert "Individual waiting times:" in lines
    sep = max(i for i, l in enumerate(lines) if l.startswith("*********"))
    csv = [l for l in lines[sep + 1:] if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    assert len(csv) == 3
    # first iterate is w=0: lsq objective = sum(y^2)/N
    X, y = data
    obj0 = float(csv[0].split(",")[1])
    expect = float((y.double() ** 2).sum() / cfg.N)
    assert abs(obj0 - expect) / max(expect, 1e-9) < 1e-4


def test_final_report_csr():
    cfg = EngineConfig(d=16, N=64, num_workers=2, objective="lsq")
    data = load_dataset(cfg, "synthetic", "synthetic", sparse=True)
    lines = _capture(cfg, _mk_res(cfg), data, sparse=True)
    sep = max(i for i, l in enumerate(lines) if l.startswith("*********"))
    csv = [l for l in lines[sep + 1:] if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    assert len(csv) == 3
    # CSR sweep must agree with densified dense sweep
    indptr, indices, values, y = data
    X = torch.zeros(cfg.N, cfg.d)
    for r in range(cfg.N):
        for j in range(int(indptr[r]), int(indptr[r + 1])):
            X[r, int(indices[j])] += float(values[j])
    dense_lines = _capture(cfg, _mk_res(cfg), (X, y), sparse=False)
    dsep = max(i for i, l in enumerate(dense_lines)
               if l.startswith("*********"))
    dcsv = [l for l in dense_lines[dsep + 1:]
            if re.match(r"^\d+,[0-9.eE+-]+$", l)]
    # same zero-iterate objective (first snapshot is w=0 in both)
    assert abs(float(csv[0].split(",")[1]) -
               float(dcsv[0].split(",")[1])) < 1e-4


def test_final_report_no_snapshots():
    cfg = EngineConfig(d=8, N=32, num_workers=2, snapshot_weights=False)
    data = load_dataset(cfg, "synthetic", "synthetic")
    res = RunResult(k=5, elapsed_ms=9, opt_vars=[], waiting_time={0: 1},
                    w=torch.zeros(cfg.d))
    lines = _capture(cfg, res, data, sparse=False)
    assert lines[-1] == "finished"
