"""Parameter server: master weights, staleness filter, update rules,
completion path, and run bookkeeping.

This is the MI355X analog of the reference's driver-side updater thread
(SparkASGDThread.scala:153-226 / SparkASAGAThread.scala:~150-230) plus the
scheduler completion callback (mergeResult, RDD.scala:1144-1165). The server
is single-writer on ``w`` — matching the reference's single updater thread —
so no atomics are needed on the weight vector."""

from __future__ import annotations

import math
import threading
import time
from typing import Dict, List, Optional, Tuple

import torch

from .. import ops
from ..core.context import ASYNCcontext, RDDPartialRes, workerState
from .config import EngineConfig
from .messages import WorkerResult


class Server:
    def __init__(self, cfg: EngineConfig, w0: Optional[torch.Tensor] = None,
                 device: Optional[torch.device] = None):
        self.cfg = cfg
        self.device = device or torch.device(cfg.device)
        self.w = (w0.clone().float().to(self.device) if w0 is not None
                  else torch.zeros(cfg.d, dtype=torch.float32,
                                   device=self.device))
        self.alpha_bar = (torch.zeros(cfg.d, dtype=torch.float32,
                                      device=self.device)
                          if cfg.algo == "asaga" else None)
        self.AC: ASYNCcontext[torch.Tensor] = ASYNCcontext()
        for p in range(cfg.num_workers):
            self.AC.STAT[p] = workerState(self.AC)
        self.k = 0                      # applied updates (reference ``k``)
        self.last_accept: Dict[int, bool] = {p: True
                                             for p in range(cfg.num_workers)}
        self.start_time = time.perf_counter()
        self.opt_vars: List[Tuple[int, torch.Tensor]] = []
        self.finish_time: Dict[int, float] = {}
        self.submit_time: Dict[int, float] = {}
        self.waiting_time: Dict[int, int] = {}
        self._clock_lock = threading.Lock()
        if cfg.snapshot_weights:
            self.opt_vars.append((0, self.w.detach().cpu().clone()))

    # -- completion path (reference mergeResult, RDD.scala:1144-1165) --------
    def on_completion(self, res: WorkerResult) -> RDDPartialRes:
        """Invoked by whoever receives a worker result (local worker thread
        or the remote-channel proxy): computes staleness against the arrival
        clock, bumps CurrentTime, refreshes STAT, and mails the result."""
        with self._clock_lock:
            now_clock = self.AC.getCurrentTime()
            staleness = now_clock - res.ts
            self.AC.add2currentTime(1)
        st = self.AC.STAT[res.worker_id]
        st.setStaleness(staleness)
        st.setAvailability(True)
        n = st.getNumTasks()
        prev = st.getAverageTaskTime()
        st.setAverageTaskTime(int((prev * n + res.elapsed_ms) / (n + 1)))
        st.updateNumTasks(1)
        # ASGD packs computed staleness (ASYNCreduce, RDD.scala:1151);
        # ASAGA packs the raw submit clock (ASYNCaggregate, RDD.scala:1333).
        ts_field = res.ts if self.cfg.algo == "asaga" else staleness
        pr = RDDPartialRes(res, ts_field, RDDPartialRes.INT_MIN, res.worker_id)
        self.AC.put(pr)
        return pr

    # -- tau filter (SparkASGDThread.scala:172 / SparkASAGAThread.scala:191) -
    def accepts(self, pr: RDDPartialRes) -> bool:
        if self.cfg.algo == "asaga":
            return (self.k - pr.getStaleness()) <= self.cfg.taw
        return pr.getStaleness() <= self.cfg.taw

    # -- update rules --------------------------------------------------------
    def apply(self, res: WorkerResult) -> None:
        """Apply one accepted gradient. ASGD: g/parRecs, step
        gamma/sqrt(k/P+1) (SparkASGDThread.scala:188-192). ASAGA: SAGA
        triple-axpy with constant step (SparkASAGAThread.scala:217-220)."""
        cfg = self.cfg
        g = res.g
        if g.device != self.device:
            g = g.to(self.device)
        if cfg.algo == "asaga":
            ops.saga_update(self.w, g, self.alpha_bar, cfg.gamma,
                            1.0 / cfg.par_recs, 1.0 / cfg.N)
        else:
            # NB: k/numPart is Scala Int division in the reference
            # (SparkASGDThread.scala:190) — keep the integer semantics.
            gamma_k = cfg.gamma / math.sqrt(self.k // cfg.num_workers + 1)
            ops.sgd_update(self.w, g, gamma_k, 1.0 / cfg.par_recs)

    def maybe_log(self) -> Optional[int]:
        """printer_freq hook: returns current k if it logged
        (reference :195-198)."""
        if self.k % self.cfg.printer_freq == 0:
            if self.cfg.snapshot_weights:
                ms = int((time.perf_counter() - self.start_time) * 1000)
                self.opt_vars.append((ms, self.w.detach().cpu().clone()))
            return self.k
        return None

    def available_workers(self) -> int:
        return self.AC.STAT[0].getAvailableWorkers()

    def elapsed_ms(self) -> int:
        return int((time.perf_counter() - self.start_time) * 1000)
