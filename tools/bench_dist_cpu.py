#!/usr/bin/env python3
"""A/B the two dist control planes on CPU (gloo): Python proxy threads
(engine/dist.py) vs the C++ server (csrc/server_dist.cpp).

Same topology, same wire protocol, same workers — only rank 0's control
plane differs, so the ratio isolates the Python/GIL overhead the C++
server removes. On GPU the same two paths run over RCCL (the C++ one is
opt-in via ASYNCAMD_DIST_ENGINE=native until round-2 multi-GPU validation).

    python tools/bench_dist_cpu.py [--world 2] [--workers 8] [--steps 300]
"""

import argparse
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.worker import Shard, Worker


def _rank_main(rank, init_file, out_file, engine, world, P, steps, warmup):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=world)
    try:
        cfg = EngineConfig(d=64, N=4000, num_workers=P,
                           num_iterations=warmup + steps + 1, gamma=0.01,
                           taw=2 ** 30, batch_rate=0.01, bucket_ratio=0.7,
                           printer_freq=1 << 30, delay_coeff=0.0, seed=42,
                           device="cpu", sync=False, algo="asgd",
                           snapshot_weights=False)
        X, y = synthetic_dense(cfg.N, cfg.d, seed=11)
        M = P // world
        workers = []
        for j in range(M):
            wid = rank * M + j
            s, t = row_shards(cfg.N, P)[wid]
            workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s,
                                             X=X[s:t], y=y[s:t]), cfg,
                                  device=torch.device("cpu")))
        marks = [warmup, warmup + steps]
        if engine == "native":
            from asyncframework_amd.engine.dist_native import NativeDistEngine
            eng = NativeDistEngine(cfg, workers, torch.device("cpu"),
                                   mark_at=marks)
            eng.run(verbose=False, max_wall_s=600)
            mk = eng.marks
        else:
            from asyncframework_amd.engine.dist import DistEngine
            deng = DistEngine(cfg, workers, torch.device("cpu"))
            if rank == 0:
                e, server, channels = deng.build_engine()
                e.verbose = False
                e.mark_at = set(marks)
                e.run(max_wall_s=600)
                mk = e.marks
                dist.barrier()
            else:
                deng.worker_loop()
                dist.barrier()
                mk = {}
        if rank == 0:
            el = mk[marks[1]] - mk[marks[0]]
            with open(out_file, "w") as f:
                f.write(str(steps / el))
    finally:
        dist.destroy_process_group()


def run_one(engine, world, P, steps, warmup):
    with tempfile.TemporaryDirectory() as td:
        init_file = os.path.join(td, "i")
        out_file = os.path.join(td, "o")
        mp.spawn(_rank_main, args=(init_file, out_file, engine, world, P,
                                   steps, warmup), nprocs=world, join=True)
        with open(out_file) as f:
            return float(f.read())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=2)
    ap.add_argument("--workers", type=int, default=8)
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--warmup", type=int, default=50)
    args = ap.parse_args()
    out = {}
    for engine in ("python", "native"):
        ups = run_one(engine, args.world, args.workers, args.steps,
                      args.warmup)
        out[engine] = round(ups, 1)
        print(f"{engine:>7}: {ups:10.1f} updates/s")
    out["speedup"] = round(out["native"] / out["python"], 2)
    print(json.dumps({"bench": "dist-control-plane-cpu",
                      "world": args.world, "workers": args.workers,
                      "steps": args.steps, **out}))


if __name__ == "__main__":
    main()
