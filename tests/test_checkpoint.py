"""Checkpoint/resume: periodic snapshots during an async run, exact-state
restore, and resumed-run continuation (a capability the reference lacks —
SURVEY §5.4)."""

import os

import torch

from asyncframework_amd import run as runner
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine import checkpoint as ckpt
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.server import Server
from asyncframework_amd.engine.worker import Shard, Worker


def _cfg(tmp_path=None, **kw):
    base = dict(d=16, N=256, num_workers=2, num_iterations=40, gamma=0.3,
                taw=1 << 30, batch_rate=0.3, bucket_ratio=0.5,
                printer_freq=1000, delay_coeff=0.0, seed=42, device="cpu",
                snapshot_weights=False)
    base.update(kw)
    return EngineConfig(**base)


def test_capture_and_restore_roundtrip():
    cfg = _cfg(algo="asaga")
    X, y = synthetic_dense(cfg.N, cfg.d, seed=1)
    workers = runner.build_dense_workers(cfg, X, y)
    server = Server(cfg, device=torch.device("cpu"))
    server.k = 17
    server.AC.setCurrentTime(23)
    server.w.normal_()
    server.alpha_bar.normal_()
    workers[0].alpha.normal_()
    state = ckpt.capture_state(server, workers)

    server2 = Server(cfg, device=torch.device("cpu"))
    workers2 = runner.build_dense_workers(cfg, X, y)
    ckpt.restore(server2, workers2, state)
    assert server2.k == 17
    assert server2.AC.getCurrentTime() == 23
    assert torch.equal(server2.w, server.w)
    assert torch.equal(server2.alpha_bar, server.alpha_bar)
    assert torch.equal(workers2[0].alpha, workers[0].alpha)


def test_save_load_file(tmp_path):
    cfg = _cfg()
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2)
    workers = runner.build_dense_workers(cfg, X, y)
    server = Server(cfg, device=torch.device("cpu"))
    server.w.normal_()
    p = str(tmp_path / "run.ckpt")
    ckpt.save_checkpoint(p, server, workers)
    state = ckpt.load_checkpoint(p)
    assert torch.equal(state["w"], server.w)
    assert state["k"] == 0


def test_periodic_checkpoint_during_run(tmp_path):
    p = str(tmp_path / "periodic.ckpt")
    cfg = _cfg(checkpoint_path=p, checkpoint_every=10, num_iterations=35)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=3)
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    assert os.path.exists(p)
    state = ckpt.load_checkpoint(p)
    assert state["k"] % 10 == 0 and state["k"] >= 10


def test_resume_continues_from_k(tmp_path):
    """A resumed server continues counting from the checkpointed k."""
    cfg = _cfg(num_iterations=20)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=4)
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, verbose=False)
    state = ckpt.capture_state(srv, workers)

    cfg2 = _cfg(num_iterations=40)
    workers2 = runner.build_dense_workers(cfg2, X, y)
    srv2 = Server(cfg2, device=torch.device("cpu"))
    ckpt.restore(srv2, workers2, state)
    assert srv2.k == res.k
    from asyncframework_amd.engine.local import AsyncEngine
    eng = AsyncEngine(cfg2, workers=workers2, server=srv2)
    eng.verbose = False
    res2 = eng.run(max_wall_s=60)
    assert res2.k >= 40


def test_sync_engine_checkpoints(tmp_path):
    """The sync engines honor checkpoint_every too (the async hook's
    counterpart; reference has no checkpointing at all)."""
    import torch

    from asyncframework_amd.data.synthetic import synthetic_dense
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    from asyncframework_amd.engine.config import EngineConfig
    from asyncframework_amd.engine.local import SyncEngine
    from asyncframework_amd.run import build_dense_workers
    ck = str(tmp_path / "sync.ckpt")
    cfg = EngineConfig(d=8, N=80, num_workers=2, num_iterations=30,
                       gamma=0.2, batch_rate=0.3, bucket_ratio=1.0,
                       printer_freq=1000, delay_coeff=0.0, seed=1,
                       device="cpu", sync=True, algo="asgd",
                       snapshot_weights=False, checkpoint_path=ck,
                       checkpoint_every=10)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2)
    eng = SyncEngine(cfg, build_dense_workers(cfg, X, y))
    eng.verbose = False
    eng.run(max_wall_s=60)
    state = load_checkpoint(ck)
    assert state["k"] % 10 == 0 and state["k"] >= 10
    assert torch.is_tensor(state["w"])


def test_sync_engine_resume_continues_from_k(tmp_path):
    import torch

    from asyncframework_amd.data.synthetic import synthetic_dense
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    from asyncframework_amd.engine.config import EngineConfig
    from asyncframework_amd.run import build_dense_workers, run_engine
    ck = str(tmp_path / "s.ckpt")
    kw = dict(d=8, N=80, num_workers=2, num_iterations=30, gamma=0.2,
              batch_rate=0.3, bucket_ratio=1.0, printer_freq=1000,
              delay_coeff=0.0, seed=1, device="cpu", sync=True, algo="asgd",
              snapshot_weights=False)
    cfg = EngineConfig(**kw, checkpoint_path=ck, checkpoint_every=10)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2)
    run_engine(cfg, build_dense_workers(cfg, X, y), verbose=False)
    state = load_checkpoint(ck)

    # resume with the budget equal to the checkpointed k: zero new rounds,
    # weights unchanged
    cfg2 = EngineConfig(**{**kw, "num_iterations": state["k"]})
    res2, _ = run_engine(cfg2, build_dense_workers(cfg2, X, y),
                         verbose=False, resume_from=ck)
    assert res2.k == state["k"]
    assert torch.equal(res2.w, state["w"])


def test_restore_rejects_mismatched_worker_pool(tmp_path):
    import pytest as _pt
    import torch

    from asyncframework_amd.data.synthetic import synthetic_dense
    from asyncframework_amd.engine.checkpoint import (load_checkpoint,
                                                      restore,
                                                      save_checkpoint)
    from asyncframework_amd.engine.config import EngineConfig
    from asyncframework_amd.engine.server import Server
    cfg = EngineConfig(d=8, N=80, num_workers=4, algo="asaga")
    srv = Server(cfg, device=torch.device("cpu"))
    p = str(tmp_path / "c.ckpt")
    save_checkpoint(p, srv)
    state = load_checkpoint(p)
    cfg2 = EngineConfig(d=8, N=80, num_workers=2, algo="asaga")
    srv2 = Server(cfg2, device=torch.device("cpu"))
    with _pt.raises(ValueError, match="num_workers"):
        restore(srv2, [], state)
    cfg3 = EngineConfig(d=16, N=80, num_workers=4, algo="asaga")
    srv3 = Server(cfg3, device=torch.device("cpu"))
    with _pt.raises(ValueError):
        restore(srv3, [], state)
