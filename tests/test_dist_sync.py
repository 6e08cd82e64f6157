"""AllReduceSyncEngine (engine/dist_sync.py) — the C6 mapping: MLlib's
treeAggregate baseline as replicated-weights all_reduce. Verified against
the single-process SyncEngine in mllib mode: same Philox masks, same update
rule => numerically matching iterates."""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.worker import Shard, Worker

WORLD = 2
P = 4


def _mk_cfg():
    return EngineConfig(d=16, N=200, num_workers=P, num_iterations=30,
                        gamma=0.5, taw=2 ** 30, batch_rate=0.3,
                        bucket_ratio=1.0, printer_freq=1000, delay_coeff=0.0,
                        seed=42, device="cpu", sync=True, algo="mllib",
                        snapshot_weights=False)


def _rank_main(rank, init_file, out_file):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=WORLD)
    try:
        from asyncframework_amd.engine.dist_sync import AllReduceSyncEngine
        cfg = _mk_cfg()
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        M = P // WORLD
        workers = []
        for j in range(M):
            wid = rank * M + j
            s, t = row_shards(cfg.N, P)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        eng = AllReduceSyncEngine(cfg, workers, torch.device("cpu"))
        res = eng.run(verbose=False, max_wall_s=60)
        if rank == 0:
            torch.save(res.w, out_file)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_allreduce_mllib_matches_single_process():
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "i")
        out_file = os.path.join(td, "o")
        mp.spawn(_rank_main, args=(init_file, out_file), nprocs=WORLD,
                 join=True)
        w_dist = torch.load(out_file, weights_only=False)

    # single-process oracle: SyncEngine in mllib mode, same P/shards/seeds
    from asyncframework_amd.engine.local import SyncEngine
    from asyncframework_amd.engine.server import Server
    from asyncframework_amd.run import build_dense_workers
    cfg = _mk_cfg()
    X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
    workers = build_dense_workers(cfg, X, y)
    eng = SyncEngine(cfg, workers, server=Server(cfg,
                                                 device=torch.device("cpu")))
    eng.verbose = False
    res = eng.run(max_wall_s=60)
    assert torch.allclose(w_dist, res.w, atol=1e-5), \
        (w_dist - res.w).abs().max()
    obj0 = float(((X @ torch.zeros(cfg.d) - y) ** 2).mean())
    obj1 = float(((X @ w_dist - y) ** 2).mean())
    assert obj1 < obj0
