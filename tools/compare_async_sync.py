#!/usr/bin/env python3
"""Async (ASGD/ASAGA) vs synchronous baseline under straggler injection —
the reference's headline experiment ("ASYNC is ~4x faster than
Spark" time-to-equal-error under the cloud long-tail delay model).

Runs the SAME config twice on one node: the bounded-staleness async engine
(quorum gate bucketRatio) and the sync engine (full barrier per round), both
with the reference's delay model (25% stragglers; long-tail 2.5-10x, normal
1.5-2.5x of calibrated avgDelay — SparkASGDThread.scala:124-141). Reports
wall-clock-to-target-loss for each and the speedup, as one JSON line.

Usage: python tools/compare_async_sync.py [--device cuda:0] [--rows 400000]
  [--cols 2000] [--workers 8] [--iters 3000] [--coeff -1] [--target-rel 0.3]
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from asyncframework_amd import run as runner  # noqa: E402
from asyncframework_amd.data.synthetic import synthetic_dense  # noqa: E402
from asyncframework_amd.engine.config import EngineConfig  # noqa: E402
from asyncframework_amd.ops import torch_ref  # noqa: E402


def loss_curve(cfg, X, y, max_wall_s):
    workers = runner.build_dense_workers(cfg, X, y)
    res, srv = runner.run_engine(cfg, workers, max_wall_s=max_wall_s,
                                 verbose=False)
    W = torch.stack([w for (_, w) in res.opt_vars]).to(X.device)
    obj = torch_ref.objective_sweep(X, y, W, cfg.objective)
    return ([(t, float(o)) for (t, _), o in zip(res.opt_vars, obj)],
            res.k, res.elapsed_ms)


def time_to_target(curve, target):
    for t_ms, o in curve:
        if o <= target:
            return t_ms
    return None


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available()
                   else "cpu")
    p.add_argument("--rows", type=int, default=400_000)   # epsilon shape
    p.add_argument("--cols", type=int, default=2_000)
    p.add_argument("--workers", type=int, default=8)
    p.add_argument("--iters", type=int, default=2000)
    p.add_argument("--gamma", type=float, default=1.0)
    p.add_argument("--rate", type=float, default=0.01)
    p.add_argument("--bucket-ratio", type=float, default=0.7)
    p.add_argument("--taw", type=int, default=20_000_000)
    p.add_argument("--coeff", type=float, default=-1.0)
    p.add_argument("--calib-factor", type=int, default=10)
    p.add_argument("--target-rel", type=float, default=0.35,
                   help="target = rel * initial objective")
    p.add_argument("--dtype", default="fp32")
    p.add_argument("--max-wall-s", type=float, default=300.0)
    args = p.parse_args()

    def mk_cfg(sync):
        # sync does one update per ROUND of P tasks; async applies P updates
        # per round-equivalent — equal-iteration counts match total task work
        iters = args.iters if not sync else max(1, args.iters // args.workers)
        return EngineConfig(
            d=args.cols, N=args.rows, num_workers=args.workers,
            num_iterations=iters, gamma=args.gamma, taw=args.taw,
            batch_rate=args.rate, bucket_ratio=args.bucket_ratio,
            printer_freq=max(1, iters // 40), delay_coeff=args.coeff,
            seed=42, algo="asgd", sync=sync, objective="lsq",
            dtype=args.dtype, device=args.device,
            calib_factor=args.calib_factor, snapshot_weights=True)

    dt = mk_cfg(False).torch_dtype()
    X, y = synthetic_dense(args.rows, args.cols, seed=42, dtype=dt,
                           device=args.device)

    async_curve, ak, ams = loss_curve(mk_cfg(False), X, y, args.max_wall_s)
    sync_curve, sk, sms = loss_curve(mk_cfg(True), X, y, args.max_wall_s)

    obj0 = async_curve[0][1]
    floor = max(min(o for _, o in async_curve),
                min(o for _, o in sync_curve))
    target = max(args.target_rel * obj0, floor * 1.02)
    ta = time_to_target(async_curve, target)
    ts = time_to_target(sync_curve, target)
    out = {
        "experiment": "async_vs_sync_time_to_target_loss",
        "delay_model": ("cloud-long-tail" if args.coeff == -1.0
                        else f"coeff={args.coeff}"),
        "config": {"rows": args.rows, "cols": args.cols,
                   "workers": args.workers, "rate": args.rate,
                   "bucket_ratio": args.bucket_ratio, "gamma": args.gamma,
                   "device": args.device},
        "initial_objective": obj0,
        "target_objective": target,
        "async": {"iters": ak, "elapsed_ms": ams,
                  "time_to_target_ms": ta,
                  "final_obj": async_curve[-1][1]},
        "sync": {"rounds": sk, "elapsed_ms": sms,
                 "time_to_target_ms": ts,
                 "final_obj": sync_curve[-1][1]},
        "speedup_to_target": (round(ts / ta, 3)
                              if (ta is not None and ts is not None and ta > 0)
                              else None),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
