"""All-reduce synchronous engine — the C6 path of SURVEY §2.7.

The reference's MLlib baseline aggregates per-iteration gradients with a
2-level treeAggregate (GradientDescent.scala:267-277). The MI355X-native
equivalent is a replicated-weights data-parallel loop over
``dist.all_reduce`` (RCCL ring over xGMI on GPU, gloo on CPU): every rank
holds an identical ``w`` replica, computes the gradient sum over its M
local workers, all-reduces (gradient, nrows), and applies the identical
update locally — no parameter-server rank, no pair channels, and the
collective IS the barrier.

This engine serves the ``mllib`` algorithm under torchrun (the async
engines and the user-space-barrier sync variants keep the mailbox
topology, which is the reference's own shape for those drivers —
SparkASGDSync counts results in user space, SparkASGDSync.scala:239-271).
"""

from __future__ import annotations

import math
import time
from typing import List, Optional

import torch
import torch.distributed as dist

from .config import EngineConfig
from .local import RunResult
from .messages import Dispatch
from .worker import Worker


class AllReduceSyncEngine:
    """Replicated-w synchronous mini-batch SGD over all_reduce."""

    def __init__(self, cfg: EngineConfig, local_workers: List[Worker],
                 device: torch.device):
        assert dist.is_initialized()
        self.cfg = cfg
        self.workers = local_workers
        self.device = device
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        assert cfg.num_workers % self.world == 0
        assert cfg.algo in ("mllib", "asgd"), "sync rules: mllib or asgd"

    def run(self, max_wall_s: Optional[float] = None,
            verbose: bool = True) -> Optional[RunResult]:
        cfg = self.cfg
        w = torch.zeros(cfg.d, dtype=torch.float32, device=self.device)
        acc = torch.zeros_like(w)
        meta = torch.zeros(1, dtype=torch.float32, device=self.device)
        opt_vars = []
        t0 = time.perf_counter()
        if self.rank == 0 and cfg.snapshot_weights:
            opt_vars.append((0, w.cpu().clone()))
        k = 0
        for k in range(cfg.num_iterations):
            if max_wall_s and time.perf_counter() - t0 > max_wall_s:
                break
            acc.zero_()
            nrows = 0
            for wk in self.workers:
                res = wk.process(Dispatch(w=w, ts=k, k_submit=k))
                g = res.g
                if g.device != self.device:
                    g = g.to(self.device)
                acc += g
                nrows += res.nrows
            meta[0] = float(nrows)
            # the collective is the barrier (RCCL ring over xGMI / gloo)
            dist.all_reduce(acc)
            dist.all_reduce(meta)
            total_rows = max(int(meta.item()), 1)
            gamma_k = cfg.gamma / math.sqrt(k + 1)
            if cfg.algo == "mllib":
                # grad sum / ACTUAL minibatch size, SimpleUpdater step
                # (reference GradientDescent.scala:287-290)
                w.add_(acc, alpha=-gamma_k / total_rows)
            else:
                # SparkASGDSync rule: / (b*N) (SparkASGDSync.scala:273-277)
                w.add_(acc, alpha=-gamma_k / (cfg.batch_rate * cfg.N))
            if self.rank == 0 and (k % cfg.printer_freq == 0):
                if verbose:
                    print(f"Iteration {k} is finished")
                if cfg.snapshot_weights:
                    ms = int((time.perf_counter() - t0) * 1000)
                    opt_vars.append((ms, w.cpu().clone()))
        elapsed = int((time.perf_counter() - t0) * 1000)
        if self.rank != 0:
            return None
        return RunResult(k=k + 1 if cfg.num_iterations else 0,
                         elapsed_ms=elapsed, opt_vars=opt_vars,
                         waiting_time={i: 0
                                       for i in range(cfg.num_workers)},
                         w=w, applied=k + 1, rejected=0)
