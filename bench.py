#!/usr/bin/env python3
"""bench.py — flagship benchmark: gradient updates/sec (whole node),
mnist8m-shape ASGD (BASELINE.json metric/config).

Contract (driver): ``python bench.py --gpus N --steps K --warmup W``.
For N>1 the driver launches one rank per GPU via torch.distributed.run; each
rank reads RANK/LOCAL_RANK/WORLD_SIZE from the env. A *step* is one applied
gradient update at the parameter server (the reference's iteration ``k``,
SparkASGDThread.scala:199). W warmup updates run untimed; then EXACTLY K
updates are timed between wall-clock marks taken by the single-writer server
thread, bracketed by a global barrier + torch.cuda.synchronize on both
sides. value = K / elapsed = whole-node applied updates/sec (all N GPUs feed
the same server). Rank 0 prints one JSON line.

Dataset: synthetic mnist8m-shape (8,100,000 x 784) dense bf16, random-init
weights (no network for real datasets — BASELINE.md). The dataset is FIXED
as N grows (sharded over workers) => strong scaling, matching the
reference's fixed-dataset/more-partitions semantics.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.data.synthetic import synthetic_dense
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.local import AsyncEngine
from asyncframework_amd.engine.server import Server
from asyncframework_amd.engine.worker import Shard, Worker

BASE = dict(rows=8_100_000, cols=784, rate=0.01, taw=20_000_000,
            gamma=1.5625e-3, bucket_ratio=0.7, seed=42)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=500)
    p.add_argument("--rows", type=int, default=BASE["rows"])
    p.add_argument("--cols", type=int, default=BASE["cols"])
    p.add_argument("--rate", type=float, default=BASE["rate"])
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--objective", default="lsq", choices=["lsq", "logistic"])
    p.add_argument("--algo", default="asgd", choices=["asgd", "asaga"])
    p.add_argument("--device", default=None, help="override (cpu for debug)")
    p.add_argument("--engine", default="graph", choices=["graph", "threads"],
                   help="N=1 GPU path: hipGraph device loop (default) or the "
                        "threaded mailbox engine")
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(args.gpus, world)

    if args.device:
        device = torch.device(args.device)
    elif torch.cuda.is_available():
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    if device.type == "cuda":
        # the native kernels are mandatory on GPU (ops dispatch enforces it)
        import asyncframework_amd.ops as ops
        assert ops.hip_available(), "HIP extension missing on a GPU box"

    cfg = EngineConfig(
        d=args.cols, N=args.rows, num_workers=world,
        num_iterations=args.warmup + args.steps + 1,
        gamma=BASE["gamma"], taw=BASE["taw"], batch_rate=args.rate,
        bucket_ratio=BASE["bucket_ratio"], printer_freq=1 << 30,
        delay_coeff=0.0, seed=BASE["seed"], algo=args.algo, sync=False,
        objective=args.objective, dtype=args.dtype,
        device=str(device), snapshot_weights=False)

    dtype = cfg.torch_dtype()
    shards = row_shards(args.rows, world)
    s, t = shards[rank]
    # every rank generates only ITS shard (seeded per-rank so ranks differ)
    X, y = synthetic_dense(t - s, args.cols, seed=BASE["seed"] + rank,
                           dtype=dtype, device=device,
                           objective=args.objective)
    worker = Worker(rank, Shard(row_start=s, n_rows=t - s, X=X, y=y), cfg,
                    device=device)

    mark_lo, mark_hi = args.warmup, args.warmup + args.steps

    if world > 1:
        import torch.distributed as dist
        from asyncframework_amd.engine.dist import DistEngine
        dist.init_process_group(
            "nccl" if device.type == "cuda" else "gloo",
            rank=rank, world_size=world)
        dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()
        deng = DistEngine(cfg, worker, device)
        if rank == 0:
            server = Server(cfg, device=device)
            from asyncframework_amd.engine.local import _LocalChannel
            channels = [_LocalChannel(worker, server)]
            from asyncframework_amd.engine.dist import _RemoteChannel
            for i in range(1, world):
                channels.append(_RemoteChannel(i, deng.pair_groups[i],
                                               server, cfg, device))
            eng = AsyncEngine(cfg, server=server, channels=channels)
            eng.verbose = False
            eng.mark_at = {mark_lo, mark_hi}
            res = eng.run(max_wall_s=1800)
            dist.barrier()
            if device.type == "cuda":
                torch.cuda.synchronize()
            emit(args, cfg, eng, n_gpus=world)
        else:
            from asyncframework_amd.engine.dist import remote_worker_loop
            remote_worker_loop(worker, cfg, deng.pair_groups[rank], device)
            dist.barrier()
            if device.type == "cuda":
                torch.cuda.synchronize()
        dist.destroy_process_group()
    elif device.type == "cuda" and args.engine == "graph":
        # GPU-resident hipGraph round loop (engine/graph.py): the whole
        # [sample+grad, fused update] round replayed from a captured graph.
        from asyncframework_amd.engine.graph import GraphEngine
        geng = GraphEngine(cfg, worker.shard, device)
        torch.cuda.synchronize()
        t0, t1 = geng.bench(args.warmup, args.steps)
        emit_elapsed(args, cfg, t1 - t0, n_gpus=1)
    else:
        if device.type == "cuda":
            torch.cuda.synchronize()
        eng = AsyncEngine(cfg, workers=[worker])
        eng.verbose = False
        eng.mark_at = {mark_lo, mark_hi}
        res = eng.run(max_wall_s=1800)
        if device.type == "cuda":
            torch.cuda.synchronize()
        emit(args, cfg, eng, n_gpus=1)


def emit(args, cfg: EngineConfig, eng: AsyncEngine, n_gpus: int):
    lo, hi = args.warmup, args.warmup + args.steps
    t0 = eng.marks.get(lo)
    t1 = eng.marks.get(hi)
    if t0 is None or t1 is None or t1 <= t0:
        print(json.dumps({"error": "marks missing", "marks":
                          {str(k): v for k, v in eng.marks.items()}}))
        sys.exit(1)
    emit_elapsed(args, cfg, t1 - t0, n_gpus)


def emit_elapsed(args, cfg: EngineConfig, elapsed: float, n_gpus: int):
    ups = args.steps / elapsed
    out = {
        "metric": "gradient updates/sec (whole node)",
        "value": round(ups, 2),
        "unit": "updates/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1000.0 / args.steps, 4),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": args.dtype,
        "data": "synthetic",
        "config": {
            "model": f"{args.algo}-{args.objective}-mnist8m",
            "rows": args.rows, "cols": args.cols,
            "batch_rate": args.rate, "taw": cfg.taw,
            "gamma": cfg.gamma, "bucket_ratio": cfg.bucket_ratio,
            "global_batch": int(args.rate * args.rows),
            "seq_len": args.cols,
            "parallelism": f"async-ps-dp{n_gpus}",
        },
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
