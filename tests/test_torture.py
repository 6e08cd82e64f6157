"""Torture mode: heavy straggler injection + tight staleness
bound. Asserts the tau contract holds under stress: every ACCEPTED update had
staleness <= taw (the reference silently drops over-tau results and requeues
the worker, SparkASGDThread.scala:172,202-205), rejects actually happen, and
the run still converges to completion."""

import torch

from asyncframework_amd import run as runner
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.delay import DelayInjector
from asyncframework_amd.engine.local import AsyncEngine
from asyncframework_amd.engine.server import Server


def test_torture_staleness_contract():
    cfg = EngineConfig(d=16, N=512, num_workers=6, num_iterations=150,
                       gamma=0.2, taw=2, batch_rate=0.2, bucket_ratio=0.3,
                       printer_freq=10 ** 9, delay_coeff=-1.0, seed=7,
                       device="cpu", snapshot_weights=False,
                       calib_factor=2)  # calibrate fast -> delays kick in
    X, y = synthetic_dense(cfg.N, cfg.d, seed=5)
    workers = runner.build_dense_workers(cfg, X, y)
    server = Server(cfg, device=torch.device("cpu"))
    delay = DelayInjector(cfg.num_workers, cfg.delay_coeff, cfg.seed,
                          calib_window=cfg.calib_factor * cfg.num_workers)
    eng = AsyncEngine(cfg, workers=workers, server=server, delay=delay)
    eng.verbose = False
    res = eng.run(max_wall_s=120)
    assert res.k >= cfg.num_iterations
    assert len(eng.accepted_staleness) >= cfg.num_iterations
    assert max(eng.accepted_staleness) <= cfg.taw
    # the injector was active and produced real delays
    assert delay.flag


def test_torture_tau_zero_many_workers():
    cfg = EngineConfig(d=8, N=256, num_workers=8, num_iterations=60,
                       gamma=0.2, taw=0, batch_rate=0.3, bucket_ratio=0.25,
                       printer_freq=10 ** 9, delay_coeff=0.0, seed=9,
                       device="cpu", snapshot_weights=False)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=6)
    workers = runner.build_dense_workers(cfg, X, y)
    eng = AsyncEngine(cfg, workers=workers,
                      server=Server(cfg, device=torch.device("cpu")))
    eng.verbose = False
    res = eng.run(max_wall_s=120)
    assert res.k >= cfg.num_iterations
    assert max(eng.accepted_staleness) <= 0


def test_torture_delay_plus_tight_tau():
    """SURVEY §5.2's torture recipe: straggler injection (cloud model,
    tiny calibration window) + tight tau together; the run must still
    reach its iteration budget and never apply an over-tau result."""
    from asyncframework_amd.data.synthetic import synthetic_dense
    from asyncframework_amd.engine.config import EngineConfig
    from asyncframework_amd.engine.local import AsyncEngine
    from asyncframework_amd.run import build_dense_workers
    cfg = EngineConfig(d=12, N=240, num_workers=6, num_iterations=60,
                       gamma=0.1, taw=2, batch_rate=0.2, bucket_ratio=0.3,
                       printer_freq=1 << 30, delay_coeff=-1.0, seed=3,
                       device="cpu", snapshot_weights=False,
                       calib_factor=2)  # calibration after 12 tasks
    X, y = synthetic_dense(cfg.N, cfg.d, seed=9)
    eng = AsyncEngine(cfg, build_dense_workers(cfg, X, y))
    eng.verbose = False
    res = eng.run(max_wall_s=120)
    assert res.k >= cfg.num_iterations
    assert max(eng.accepted_staleness) <= cfg.taw
    assert eng.delay.flag  # calibration really activated mid-run


def test_concurrent_context_hammer():
    """ASYNCcontext under concurrent producers + clock bumps: no lost
    results, monotone clock (the mailbox is the one synchronized point,
    as in the reference's JobWaiter.taskSucceeded)."""
    import threading

    from asyncframework_amd.core.context import ASYNCcontext, RDDPartialRes
    AC = ASYNCcontext()
    NP, PER = 8, 500

    def producer(pid):
        for i in range(PER):
            AC.put(RDDPartialRes(data=(pid, i), ts=0, id=pid))
            AC.add2currentTime(1)

    threads = [threading.Thread(target=producer, args=(p,))
               for p in range(NP)]
    for t in threads:
        t.start()
    seen = set()
    for _ in range(NP * PER):
        pr = AC.ASYNCcollectAll(timeout=30)
        seen.add(pr.gettaskResult())
    for t in threads:
        t.join()
    assert len(seen) == NP * PER
    assert AC.getCurrentTime() == NP * PER
    assert not AC.hasNext()
