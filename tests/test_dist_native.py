"""The C++ dist server (csrc/server_dist.cpp) over gloo on CPU, world=2 —
the SAME C++ loop that drives RCCL on GPU (only the ProcessGroup backend
differs). Mirrors tests/test_engine_dist.py's scenarios."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.dist_native import (NativeDistEngine,
                                                   dist_core_available)
from asyncframework_amd.engine.worker import Shard, Worker

pytestmark = pytest.mark.skipif(not dist_core_available(),
                                reason="_dist_core.so not built")

WORLD = 2


def _rank_main(rank, init_file, out_file, algo, P):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        cfg = EngineConfig(d=24, N=400, num_workers=P, num_iterations=60,
                           gamma=0.5 if algo == "asgd" else 0.05,
                           taw=2 ** 30, batch_rate=0.3, bucket_ratio=0.5,
                           printer_freq=20, delay_coeff=0.0, seed=42,
                           device="cpu", sync=False, algo=algo)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        M = P // WORLD
        workers = []
        for j in range(M):
            wid = rank * M + j
            s, t = row_shards(cfg.N, P)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"),
                               mark_at=[10, 50])
        res = eng.run(verbose=False, max_wall_s=120)
        if rank == 0:
            obj0 = float(((X @ torch.zeros(cfg.d) - y) ** 2).mean())
            obj1 = float(((X @ res.w - y) ** 2).mean())
            n_snap = len(res.opt_vars)
            n_marks = len(eng.marks)
            n_wait = sum(1 for v in res.waiting_time.values() if v >= 0)
            with open(out_file, "w") as f:
                f.write(f"{res.k},{obj0},{obj1},{res.applied},"
                        f"{n_snap},{n_marks},{n_wait}")
    finally:
        dist.destroy_process_group()


def _run(algo, P):
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "init")
        out_file = os.path.join(td, "out")
        mp.spawn(_rank_main, args=(init_file, out_file, algo, P),
                 nprocs=WORLD, join=True)
        with open(out_file) as f:
            return f.read().split(",")


@pytest.mark.timeout(300)
def test_native_dist_asgd_converges():
    k, obj0, obj1, applied, n_snap, n_marks, n_wait = _run("asgd", P=2)
    assert int(k) >= 60
    assert float(obj1) < float(obj0)
    assert int(applied) >= 60
    assert int(n_snap) >= 3    # 0 + every printer_freq=20
    assert int(n_marks) == 2   # both bench marks stamped
    assert int(n_wait) == 2


@pytest.mark.timeout(300)
def test_native_dist_asaga_converges():
    k, obj0, obj1, applied, *_ = _run("asaga", P=2)
    assert int(k) >= 60
    assert float(obj1) < float(obj0)


@pytest.mark.timeout(300)
def test_native_dist_multi_worker_per_rank():
    k, obj0, obj1, applied, n_snap, n_marks, n_wait = _run("asgd", P=4)
    assert int(k) >= 60
    assert float(obj1) < float(obj0)
    assert int(n_wait) == 4    # every logical worker dispatched


def test_delay_model_matches_python_injector():
    """The C++ straggler model is bit-compatible with engine/delay.py
    (same Philox draws, same selection/calibration rules)."""
    from asyncframework_amd.engine.delay import DelayInjector
    from asyncframework_amd.engine.dist_native import _load
    core = _load()
    for P in (4, 8, 32):
        for coeff in (-1.0, 0.0, 1.0, 2.5):
            inj = DelayInjector(P, coeff=coeff, seed=42, calib_window=0)
            inj._cul_time, inj._cul_count = 100.0, 1
            inj.maybe_activate(1)
            for wid in range(P):
                for rk in (0, 3, 17):
                    assert core.DistServer.delay_probe(
                        P, coeff, 42, 100.0, wid, rk) == \
                        inj.delay_ms(wid, rk), (P, coeff, wid, rk)


def _tau_rank_main(rank, init_file, out_file, algo):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        # taw=0: only zero-staleness results accepted — heavy rejection
        # traffic; the server must still reach num_iterations (the
        # reference silently drops over-tau results and requeues,
        # SparkASGDThread.scala:202-205)
        cfg = EngineConfig(d=12, N=160, num_workers=4, num_iterations=25,
                           gamma=0.2 if algo == "asgd" else 0.02, taw=0,
                           batch_rate=0.3, bucket_ratio=0.25,
                           printer_freq=1000, delay_coeff=0.0, seed=7,
                           device="cpu", sync=False, algo=algo,
                           snapshot_weights=False)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=3)
        M = 2
        workers = []
        for j in range(M):
            wid = rank * M + j
            s, t = row_shards(cfg.N, 4)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=60)
        if rank == 0:
            with open(out_file, "w") as f:
                f.write(f"{res.k},{res.applied},{res.rejected}")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("algo", ["asgd", "asaga"])
def test_native_dist_tau_zero_torture(algo):
    """taw=0 under both staleness encodings (ASGD arrival-clock,
    ASAGA k-ts): heavy rejection traffic, run still completes."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "init")
        out_file = os.path.join(td, "out")
        mp.spawn(_tau_rank_main, args=(init_file, out_file, algo),
                 nprocs=WORLD, join=True)
        with open(out_file) as f:
            k, applied, rejected = map(int, f.read().split(","))
        assert k >= 25
        assert applied >= 25
        # with taw=0 and 4 concurrent workers rejections must occur
        assert rejected > 0


def _wall_rank_main(rank, init_file, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        # unreachable iteration budget: the wall-clock cap must stop the
        # run cleanly on every rank (no hang, partial results returned)
        cfg = EngineConfig(d=12, N=160, num_workers=2, num_iterations=10 ** 9,
                           gamma=0.1, taw=2 ** 30, batch_rate=0.3,
                           bucket_ratio=0.5, printer_freq=1 << 30,
                           delay_coeff=0.0, seed=7, device="cpu",
                           sync=False, algo="asgd", snapshot_weights=False)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=3)
        s, t = row_shards(cfg.N, 2)[rank]
        worker = Worker(rank, Shard(row_start=s, n_rows=t - s, X=X[s:t],
                                    y=y[s:t]), cfg,
                        device=torch.device("cpu"))
        eng = NativeDistEngine(cfg, [worker], torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=2.0)
        if rank == 0:
            with open(out_file, "w") as f:
                f.write(f"{res.k},{res.elapsed_ms}")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_wall_clock_cap():
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "init")
        out_file = os.path.join(td, "out")
        mp.spawn(_wall_rank_main, args=(init_file, out_file), nprocs=WORLD,
                 join=True)
        with open(out_file) as f:
            k, elapsed = map(int, f.read().split(","))
        assert k > 0          # made progress
        assert elapsed < 30_000  # stopped near the cap, not the timeout


def _delay_rank_main(rank, init_file, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        # coeff=5.0 + tiny calibration window: once active, worker 0's
        # rounds are delayed by 5x the measured average round time — the
        # end-to-end wiring (C++ server computes delay_s -> header ->
        # Python worker sleeps) must slow the run measurably
        cfg = EngineConfig(d=12, N=160, num_workers=2, num_iterations=40,
                           gamma=0.1, taw=2 ** 30, batch_rate=0.3,
                           bucket_ratio=0.5, printer_freq=1 << 30,
                           delay_coeff=5.0, seed=7, device="cpu",
                           sync=False, algo="asgd", snapshot_weights=False,
                           calib_factor=2)  # window = 2*P = 4 tasks
        X, y = synthetic_dense(cfg.N, cfg.d, seed=3)
        s, t = row_shards(cfg.N, 2)[rank]
        worker = Worker(rank, Shard(row_start=s, n_rows=t - s, X=X[s:t],
                                    y=y[s:t]), cfg,
                        device=torch.device("cpu"))
        eng = NativeDistEngine(cfg, [worker], torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=60)
        if rank == 0:
            srv = eng.srv
            with open(out_file, "w") as f:
                f.write(f"{res.k},{int(srv.delay_active())},"
                        f"{srv.avg_delay_ms()},{res.waiting_time[1]}")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_delay_injection_active():
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "init")
        out_file = os.path.join(td, "out")
        mp.spawn(_delay_rank_main, args=(init_file, out_file), nprocs=WORLD,
                 join=True)
        with open(out_file) as f:
            k, active, avg_ms, wait1 = f.read().split(",")
        assert int(k) >= 40
        assert int(active) == 1          # calibration completed
        assert float(avg_ms) > 0.0
        # worker 1 (not a straggler) accumulates waiting time while the
        # server's round cadence is held back by straggling worker 0
        assert int(wait1) >= 0


def _world1_run(device: str):
    """world=1: no remote channels — the C++ server drives M local Python
    workers through the slot API only. On GPU this validates the server's
    aten-update path + GIL-released slot handoff with cuda tensors on a
    single device (the multi-GPU channel path shares all of this code)."""
    import tempfile as _tf
    # GPU boxes may not resolve their own hostname; pin gloo to loopback
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    with _tf.TemporaryDirectory() as td:
        dist.init_process_group("gloo", init_method=f"file://{td}/i",
                                rank=0, world_size=1)
        try:
            dev = torch.device(device)
            cfg = EngineConfig(d=32, N=400, num_workers=4,
                               num_iterations=80, gamma=0.5, taw=2 ** 30,
                               batch_rate=0.2, bucket_ratio=0.5,
                               printer_freq=40, delay_coeff=0.0, seed=42,
                               device=device, sync=False, algo="asgd")
            X, y = synthetic_dense(cfg.N, cfg.d, seed=11, device=dev)
            workers = []
            for wid in range(4):
                s, t = row_shards(cfg.N, 4)[wid]
                workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                                 X=X[s:t], y=y[s:t]), cfg,
                                      device=dev))
            eng = NativeDistEngine(cfg, workers, dev, mark_at=[20, 70])
            res = eng.run(verbose=False, max_wall_s=120)
            obj0 = float(((X.float() @ torch.zeros(cfg.d, device=dev) - y)
                          ** 2).mean())
            obj1 = float(((X.float() @ res.w - y) ** 2).mean())
            assert res.k >= 80
            assert obj1 < obj0
            assert len(eng.marks) == 2
            assert len(res.opt_vars) >= 2
        finally:
            dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_world1_cpu():
    _world1_run("cpu")


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_native_dist_world1_gpu():
    _world1_run("cuda:0")


def _ckpt_rank_main(rank, init_file, ck_path, out_file, resume):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        cfg = EngineConfig(d=24, N=400, num_workers=4,
                           num_iterations=3000, gamma=0.05, taw=2 ** 30,
                           batch_rate=0.3, bucket_ratio=0.5,
                           printer_freq=1 << 30, delay_coeff=0.0, seed=42,
                           device="cpu", sync=False, algo="asaga",
                           snapshot_weights=False)
        if resume:
            from asyncframework_amd.engine.checkpoint import load_checkpoint
            cfg.num_iterations = load_checkpoint(ck_path)["k"]
        else:
            cfg.checkpoint_path = ck_path
            cfg.checkpoint_every = 500
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        M = 2
        workers = []
        for j in range(M):
            wid = rank * M + j
            s, t = row_shards(cfg.N, 4)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=120,
                      resume_from=ck_path if resume else "")
        if resume and rank == 1:
            torch.save([w.alpha.clone() for w in workers],
                       out_file + ".alpha1")
        if rank == 0:
            torch.save({"k": res.k, "w": res.w.clone()}, out_file)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_checkpoint_and_resume(tmp_path):
    """The C++ snap sideband: remote SAGA tables land in mid-run
    checkpoints; resume pushes everything back (checkpoints share the
    threads-engine schema)."""
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    ck = str(tmp_path / "n.ckpt")
    out1 = str(tmp_path / "o1")
    mp.spawn(_ckpt_rank_main, args=(str(tmp_path / "i1"), ck, out1, False),
             nprocs=WORLD, join=True)
    state = load_checkpoint(ck)
    assert set(state["alpha"].keys()) == {0, 1, 2, 3}
    for wid, (s, t) in enumerate(row_shards(400, 4)):
        assert state["alpha"][wid].shape == (t - s,)
    assert float(sum(a.abs().sum() for a in state["alpha"].values())) > 0
    # the native monitor snapshots asynchronously at-or-after each
    # checkpoint_every boundary (k keeps advancing during the gather),
    # unlike the threads engine's exact-multiple synchronous snapshots
    assert state["k"] >= 500
    assert state["current_time"] >= state["k"]
    assert state["alpha_bar"] is not None

    out2 = str(tmp_path / "o2")
    mp.spawn(_ckpt_rank_main, args=(str(tmp_path / "i2"), ck, out2, True),
             nprocs=WORLD, join=True)
    r = torch.load(out2, weights_only=False)
    # resumed with budget == checkpoint k: no further updates applied
    assert r["k"] == state["k"]
    assert torch.allclose(r["w"], state["w"])
    alpha1 = torch.load(out2 + ".alpha1", weights_only=False)
    assert torch.equal(alpha1[0], state["alpha"][2])
    assert torch.equal(alpha1[1], state["alpha"][3])


def _faildet_rank_main(rank, init_file, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    # bucket_ratio=1.0: the gate demands ALL FOUR workers, so once wid 3
    # goes silent every dispatch stalls — only the reaper (worker_timeout_s)
    # can unblock the run by shrinking the alive pool
    cfg = EngineConfig(d=16, N=400, num_workers=4, num_iterations=60,
                       gamma=0.3, taw=2 ** 30, batch_rate=0.3,
                       bucket_ratio=1.0, printer_freq=1 << 30,
                       delay_coeff=0.0, seed=42, device="cpu", sync=False,
                       algo="asgd", snapshot_weights=False,
                       worker_timeout_s=0.5)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
    M = 2
    workers = []
    for j in range(M):
        wid = rank * M + j
        s, t = row_shards(cfg.N, 4)[wid]
        workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                         X=X[s:t], y=y[s:t]), cfg,
                              device=torch.device("cpu")))
    if rank == 0:
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=60)
        with open(out_file, "w") as f:
            f.write(f"{res.k},{eng.srv.dead_workers()}")
        # do NOT destroy the pg: the dead peer's channel never drains and
        # destroy could block; process teardown cleans up (failure path)
        os._exit(0)
    else:
        # worker rank with a BLACK-HOLE worker: wid 3 accepts one dispatch
        # and never replies (the lost-task scenario the reference leaves
        # hanging forever)
        import threading as _th
        from asyncframework_amd.engine.dist import (_recv,
                                                    remote_worker_loop)
        from asyncframework_amd.engine.messages import HDR
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))

        def black_hole():
            buf = torch.zeros(cfg.d + HDR, dtype=torch.float32)
            _recv(buf, 0, eng.base.pair_groups[3])
            import time as _t
            _t.sleep(3600)

        th_dead = _th.Thread(target=black_hole, daemon=True)
        th_dead.start()
        remote_worker_loop(workers[0], cfg, eng.base.pair_groups[2],
                           torch.device("cpu"))
        dist.barrier()  # meet rank 0's end-of-run barrier
        os._exit(0)


@pytest.mark.timeout(300)
def test_native_dist_failure_detection():
    """A lost remote worker is declared dead after worker_timeout_s and
    excluded from the quorum gate; the run completes on the survivors."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "init")
        out_file = os.path.join(td, "out")
        mp.spawn(_faildet_rank_main, args=(init_file, out_file),
                 nprocs=WORLD, join=True)
        with open(out_file) as f:
            k, dead = map(int, f.read().split(","))
        assert k >= 60
        assert dead == 1


@pytest.mark.timeout(300)
def test_native_dist_trace_events(tmp_path):
    """With ASYNCAMD_TRACE active, the C++ server's dispatch/accept events
    are merged into the Perfetto log (world=1, all-local)."""
    import json

    from asyncframework_amd.utils import trace
    p = str(tmp_path / "nd_trace.json")
    trace.start_trace(p)
    try:
        _world1_run("cpu")
    finally:
        trace.stop_trace()
    with open(p) as f:
        evs = json.load(f)["traceEvents"]
    names = {e["name"] for e in evs}
    assert {"dispatch", "accept"} <= names
    accepts = [e for e in evs if e["name"] == "accept"]
    assert len(accepts) >= 80
    assert all(e["args"]["staleness"] >= 0 for e in accepts)
    # worker-side spans come from the Python Worker hook on the same clock
    rounds = [e for e in evs if e["name"] == "round"]
    assert len(rounds) >= 80
    # the merged clocks must interleave sensibly (same monotonic base)
    ats = sorted(e["ts"] for e in accepts)
    rts = sorted(e["ts"] for e in rounds)
    assert abs(ats[0] - rts[0]) < 60_000_000  # within a minute


def _seq_equiv_run(algo: str):
    """P=1, world=1: both the C++ server and the Python threads engine are
    strictly sequential (staleness 0, every result accepted) with identical
    Philox masks and identical update formulas — final w must be
    bit-identical to the oracle."""
    import tempfile as _tf
    cfg_kw = dict(d=20, N=200, num_workers=1, num_iterations=50,
                  gamma=0.4 if algo == "asgd" else 0.05, taw=2 ** 30,
                  batch_rate=0.3, bucket_ratio=0.5, printer_freq=1 << 30,
                  delay_coeff=0.0, seed=42, device="cpu", sync=False,
                  algo=algo, snapshot_weights=False)
    X, y = synthetic_dense(200, 20, seed=7)

    def mk_worker(cfg):
        return Worker(0, Shard(row_start=0, n_rows=200, X=X, y=y), cfg,
                      device=torch.device("cpu"))

    # oracle: threads engine
    from asyncframework_amd.engine.local import AsyncEngine
    cfg1 = EngineConfig(**cfg_kw)
    eng1 = AsyncEngine(cfg1, [mk_worker(cfg1)])
    eng1.verbose = False
    res1 = eng1.run(max_wall_s=60)

    # C++ server, all-local
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
    with _tf.TemporaryDirectory() as td:
        dist.init_process_group("gloo", init_method=f"file://{td}/i",
                                rank=0, world_size=1)
        try:
            cfg2 = EngineConfig(**cfg_kw)
            eng2 = NativeDistEngine(cfg2, [mk_worker(cfg2)],
                                    torch.device("cpu"))
            res2 = eng2.run(verbose=False, max_wall_s=60)
        finally:
            dist.destroy_process_group()

    assert res1.k == res2.k == 50
    assert torch.equal(res1.w, res2.w), \
        float((res1.w - res2.w).abs().max())


@pytest.mark.timeout(300)
def test_native_dist_bitwise_matches_threads_oracle_asgd():
    _seq_equiv_run("asgd")


@pytest.mark.timeout(300)
def test_native_dist_bitwise_matches_threads_oracle_asaga():
    _seq_equiv_run("asaga")


def _combo_rank_main(rank, init_file, ck_path, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        # everything at once: cloud stragglers + tight-ish tau + ASAGA +
        # periodic checkpoints + multi-worker-per-rank
        cfg = EngineConfig(d=16, N=400, num_workers=4, num_iterations=800,
                           gamma=0.02, taw=3, batch_rate=0.2,
                           bucket_ratio=0.5, printer_freq=1 << 30,
                           delay_coeff=-1.0, seed=5, device="cpu",
                           sync=False, algo="asaga", snapshot_weights=False,
                           calib_factor=5, checkpoint_path=ck_path,
                           checkpoint_every=200)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        workers = []
        for j in range(2):
            wid = rank * 2 + j
            s, t = row_shards(cfg.N, 4)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=120)
        if rank == 0:
            with open(out_file, "w") as f:
                f.write(f"{res.k},{res.rejected},"
                        f"{int(eng.srv.delay_active())},"
                        f"{eng.srv.max_staleness_seen()}")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_combo_delay_tau_checkpoint(tmp_path):
    from asyncframework_amd.engine.checkpoint import load_checkpoint
    init_file = str(tmp_path / "i")
    ck = str(tmp_path / "c.ckpt")
    out_file = str(tmp_path / "o")
    mp.spawn(_combo_rank_main, args=(init_file, ck, out_file), nprocs=WORLD,
             join=True)
    with open(out_file) as f:
        k, rejected, delay_on, max_stale = map(int, f.read().split(","))
    assert k >= 800
    assert delay_on == 1
    state = load_checkpoint(ck)
    assert set(state["alpha"].keys()) == {0, 1, 2, 3}
    assert state["k"] >= 200


def _w3_rank_main(rank, init_file, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=3)
    try:
        P = 6
        cfg = EngineConfig(d=20, N=396, num_workers=P, num_iterations=80,
                           gamma=0.4, taw=2 ** 30, batch_rate=0.3,
                           bucket_ratio=0.5, printer_freq=1 << 30,
                           delay_coeff=0.0, seed=42, device="cpu",
                           sync=False, algo="asgd", snapshot_weights=False)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        workers = []
        for j in range(2):
            wid = rank * 2 + j
            s, t = row_shards(cfg.N, P)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = NativeDistEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=120)
        if rank == 0:
            obj0 = float(((X @ torch.zeros(cfg.d) - y) ** 2).mean())
            obj1 = float(((X @ res.w - y) ** 2).mean())
            with open(out_file, "w") as f:
                f.write(f"{res.k},{obj0},{obj1},{len(res.waiting_time)}")
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_native_dist_three_ranks():
    """3 ranks x 2 workers: multi-peer channel fan-out (two distinct peer
    ranks served concurrently by the C++ server)."""
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "i")
        out_file = os.path.join(td, "o")
        mp.spawn(_w3_rank_main, args=(init_file, out_file), nprocs=3,
                 join=True)
        with open(out_file) as f:
            k, obj0, obj1, n_wids = f.read().split(",")
        assert int(k) >= 80
        assert float(obj1) < float(obj0)
        assert int(n_wids) == 6
