"""Worker-failure detection: a hung worker is declared dead after
worker_timeout_s, the quorum gate shrinks to the alive set, and the run
completes (the reference leaves a lost worker permanently busy and can
stall the gate — SURVEY §5.3)."""

import time

import torch

from asyncframework_amd import run as runner
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.local import AsyncEngine
from asyncframework_amd.engine.server import Server
from asyncframework_amd.engine.worker import Worker


class HangingWorker(Worker):
    """Hangs forever on its 3rd round (simulated lost executor)."""

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self._rounds = 0

    def process(self, msg):
        self._rounds += 1
        if self._rounds == 3:
            time.sleep(3600)
        return super().process(msg)


def test_hung_worker_is_reaped():
    cfg = EngineConfig(d=16, N=256, num_workers=4, num_iterations=600,
                       gamma=0.2, taw=1 << 30, batch_rate=0.3,
                       bucket_ratio=0.9,  # strict quorum: would stall
                       printer_freq=10 ** 9, delay_coeff=0.0, seed=3,
                       device="cpu", snapshot_weights=False,
                       worker_timeout_s=0.15)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=1)
    workers = runner.build_dense_workers(cfg, X, y)
    sh = workers[1].shard
    workers[1] = HangingWorker(1, sh, cfg, device=torch.device("cpu"))
    eng = AsyncEngine(cfg, workers=workers,
                      server=Server(cfg, device=torch.device("cpu")))
    eng.verbose = False
    res = eng.run(max_wall_s=60)
    assert res.k >= cfg.num_iterations, "engine stalled on the hung worker"
    assert 1 in eng.dead


def test_no_reaping_when_disabled():
    cfg = EngineConfig(d=16, N=256, num_workers=2, num_iterations=30,
                       gamma=0.2, taw=1 << 30, batch_rate=0.3,
                       bucket_ratio=0.5, printer_freq=10 ** 9,
                       delay_coeff=0.0, seed=4, device="cpu",
                       snapshot_weights=False, worker_timeout_s=0.0)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2)
    workers = runner.build_dense_workers(cfg, X, y)
    eng = AsyncEngine(cfg, workers=workers,
                      server=Server(cfg, device=torch.device("cpu")))
    eng.verbose = False
    res = eng.run(max_wall_s=60)
    assert res.k >= cfg.num_iterations
    assert not eng.dead
