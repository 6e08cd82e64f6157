"""Trace-event observability (utils/trace.py) — the MI355X analog of the
reference's event-log JSON (EventLoggingListener.scala:55): engines emit
dispatch/round/accept/reject events in chrome://tracing format."""

import json

import pytest
import torch

from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.local import AsyncEngine
from asyncframework_amd.run import build_dense_workers
from asyncframework_amd.utils import trace


@pytest.fixture(autouse=True)
def _clean_tracer(monkeypatch):
    monkeypatch.delenv("ASYNCAMD_TRACE", raising=False)
    trace.stop_trace()
    yield
    trace.stop_trace()


def test_tracer_disabled_by_default():
    assert trace.get_tracer() is None


def test_env_var_activates(tmp_path, monkeypatch):
    p = str(tmp_path / "t.json")
    monkeypatch.setenv("ASYNCAMD_TRACE", p)
    tr = trace.get_tracer()
    assert tr is not None and tr.path == p


def test_engine_run_writes_perfetto_json(tmp_path):
    p = str(tmp_path / "run_trace.json")
    trace.start_trace(p)
    cfg = EngineConfig(d=16, N=200, num_workers=3, num_iterations=30,
                       gamma=0.3, taw=2 ** 30, batch_rate=0.3,
                       bucket_ratio=0.5, printer_freq=1000, delay_coeff=0.0,
                       seed=1, device="cpu", snapshot_weights=False)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=5)
    eng = AsyncEngine(cfg, build_dense_workers(cfg, X, y))
    eng.verbose = False
    res = eng.run(max_wall_s=60)
    assert res.k >= 30
    # engine shutdown flushed the trace
    assert trace.get_tracer() is None
    with open(p) as f:
        doc = json.load(f)
    evs = doc["traceEvents"]
    names = {e["name"] for e in evs}
    assert {"dispatch", "round", "accept"} <= names
    rounds = [e for e in evs if e["name"] == "round"]
    assert all(e["ph"] == "X" and e["dur"] >= 0 for e in rounds)
    assert {e["tid"] for e in rounds} == {0, 1, 2}
    accepts = [e for e in evs if e["name"] == "accept"]
    assert len(accepts) >= 30
    assert all(e["args"]["staleness"] <= cfg.taw for e in accepts)
    # timestamps are monotone enough to be plottable
    assert all(e["ts"] >= 0 for e in evs)


def test_trace_off_costs_nothing_and_engine_still_runs():
    cfg = EngineConfig(d=8, N=80, num_workers=2, num_iterations=10,
                       gamma=0.3, taw=2 ** 30, batch_rate=0.3,
                       bucket_ratio=0.5, printer_freq=1000, delay_coeff=0.0,
                       seed=1, device="cpu", snapshot_weights=False)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=5)
    eng = AsyncEngine(cfg, build_dense_workers(cfg, X, y))
    eng.verbose = False
    assert eng.run(max_wall_s=60).k >= 10


def test_tracer_event_cap():
    tr = trace.Tracer(max_events=5)
    for i in range(9):
        tr.instant("e", 0)
    assert len(tr._events) == 5 and tr.dropped == 4
    import json
    import tempfile

    with tempfile.NamedTemporaryFile(suffix=".json") as f:
        tr.save(f.name)
        doc = json.load(open(f.name))
    assert doc["droppedEvents"] == 4
    assert len(doc["traceEvents"]) == 5


def test_stop_trace_inactive_returns_none():
    assert trace.stop_trace() is None


def test_start_trace_replaces_active(tmp_path):
    t1 = trace.start_trace(str(tmp_path / "a.json"))
    t2 = trace.start_trace(str(tmp_path / "b.json"))
    assert trace.get_tracer() is t2 and t1 is not t2
    p = trace.stop_trace()
    assert p and p.endswith("b.json")
    assert trace.get_tracer() is None
