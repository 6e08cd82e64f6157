"""Worker runtime: owns a row shard (dense or CSR) + optional SAGA history
table, and turns a Dispatch into a WorkerResult.

MI355X mapping: the reference's Executor/TaskRunner closure
pipeline becomes a per-GPU worker object whose ``process`` runs the fused HIP
gradient kernel on its own HIP stream. The SAGA history table lives with the
worker (HBM-resident, or pinned host DRAM for the spill config) instead of on
the driver (reference keeps ScalarMap driver-side,
SparkASAGAThread.scala:121; worker-resident is the 288-GB-HBM-native design
— scalars never cross the wire, only the d-dim gradient does)."""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Optional

import torch

from .. import ops
from .config import EngineConfig
from .messages import Dispatch, WorkerResult


@dataclass
class Shard:
    """A worker's contiguous row range. Dense: X [n,d]; sparse: CSR triple."""
    row_start: int
    n_rows: int
    X: Optional[torch.Tensor] = None
    y: Optional[torch.Tensor] = None
    indptr: Optional[torch.Tensor] = None
    indices: Optional[torch.Tensor] = None
    values: Optional[torch.Tensor] = None

    @property
    def is_sparse(self) -> bool:
        return self.indptr is not None


class Worker:
    """One logical worker (= one reference partition). On GPU each worker
    gets its own HIP stream so co-located workers overlap."""

    def __init__(self, worker_id: int, shard: Shard, cfg: EngineConfig,
                 device: Optional[torch.device] = None):
        self.id = worker_id
        self.shard = shard
        self.cfg = cfg
        self.device = device or (shard.X.device if shard.X is not None
                                 else shard.values.device)
        self.is_cuda = self.device.type == "cuda"
        self.stream = torch.cuda.Stream(self.device) if self.is_cuda else None
        self.alpha: Optional[torch.Tensor] = None
        self._pending_idx: Optional[torch.Tensor] = None
        self._pending_e: Optional[torch.Tensor] = None
        self._g_buf = torch.zeros(cfg.d, dtype=torch.float32, device=self.device)
        if cfg.algo == "asaga":
            self._init_history()

    def _init_history(self) -> None:
        """SAGA per-sample history scalars alpha_i for this shard
        (reference ScalarMap, SparkASAGAThread.scala:121 — one fp32 per
        sample, rank-1 reconstruction alpha_i*x_i). 'host' placement pins
        the table in host DRAM (BASELINE config 5 spill path)."""
        n = self.shard.n_rows
        if self.cfg.history_placement == "host" and self.is_cuda:
            self.alpha = torch.zeros(n, dtype=torch.float32,
                                     device="cpu").pin_memory()
            # device staging: stale except at the round's sampled rows,
            # which spill_refresh re-gathers from the pinned master each
            # round (SURVEY §7.3 — only ~rate of the table moves per round)
            self._alpha_dev_buf = torch.zeros(n, dtype=torch.float32,
                                              device=self.device)
            rate = self.cfg.batch_rate
            cap = n if rate >= 1.0 else min(n, int(rate * n * 2) + 4096)
            self._spill_cap = cap
            self._spill_rows = torch.empty(cap, dtype=torch.int32,
                                           device=self.device)
            self._spill_y = torch.empty(cap, dtype=torch.float32,
                                        device=self.device)
            self._spill_cnt = torch.zeros(1, dtype=torch.int32,
                                          device=self.device)
        else:
            self.alpha = torch.zeros(n, dtype=torch.float32, device=self.device)

    def _alpha_device(self) -> torch.Tensor:
        """History table view usable by the kernel. Host-pinned tables are
        gathered/scattered via hipMemcpyAsync-backed index ops."""
        return self.alpha

    def process(self, msg: Dispatch) -> WorkerResult:
        """Run one round: injected delay -> (SAGA) commit staged history ->
        fused sample+gradient kernel. Mirrors the reference task body
        (delay mapPartitions :287-312, sample :314, gradfun map :319-321,
        reducePartition fold RDD.scala:1103-1123)."""
        t0 = time.perf_counter()
        if msg.delay_s > 0:
            time.sleep(msg.delay_s)
        cfg = self.cfg
        sh = self.shard
        round_key = msg.k_submit + 1  # reference seed+k+1
        ctx = torch.cuda.stream(self.stream) if self.is_cuda else _nullctx()
        with ctx:
            if self.is_cuda:
                # msg.w was snapshotted on the dispatcher's stream (engine
                # thread / C++ server): order this worker's reads after it
                self.stream.wait_stream(
                    torch.cuda.default_stream(self.device))
            w = msg.w
            if w.device != self.device:
                w = w.to(self.device, non_blocking=True)
            if cfg.algo == "asaga":
                if self._pending_idx is not None and msg.accept_prev:
                    self._commit_pending()
                self._pending_idx = None
                self._pending_e = None
                g, idx, e, n = self._saga_grad(w, round_key)
                self._pending_idx, self._pending_e = idx, e
            else:
                g, n = self._grad(w, round_key)
            if self.is_cuda:
                self.stream.synchronize()
        elapsed_ms = (time.perf_counter() - t0) * 1000.0
        from ..utils.trace import get_tracer
        tr = get_tracer()
        if tr is not None:
            tr.complete("round", self.id, tr.now_us() - elapsed_ms * 1000.0,
                        elapsed_ms * 1000.0,
                        args={"k_submit": msg.k_submit, "n": n,
                              "delay_s": msg.delay_s})
        return WorkerResult(worker_id=self.id, g=g, ts=msg.ts,
                            k_submit=msg.k_submit, nrows=n,
                            elapsed_ms=elapsed_ms)

    # -- gradient dispatch ---------------------------------------------------
    def _grad(self, w, round_key):
        cfg, sh = self.cfg, self.shard
        if sh.is_sparse:
            return ops.grad_csr(sh.indptr, sh.indices, sh.values, sh.y, w,
                                seed=cfg.seed, round_k=round_key,
                                row_start=sh.row_start, rate=cfg.batch_rate,
                                objective=cfg.objective, out=self._g_buf)
        return ops.grad_dense(sh.X, sh.y, w, seed=cfg.seed, round_k=round_key,
                              row_start=sh.row_start, rate=cfg.batch_rate,
                              objective=cfg.objective, out=self._g_buf)

    def _saga_grad(self, w, round_key):
        cfg, sh = self.cfg, self.shard
        alpha = self._alpha_device()
        host_spill = alpha.device.type == "cpu" and self.is_cuda
        if host_spill:
            # Spill path (BASELINE config 5): the master table stays pinned
            # in host DRAM; refresh ONLY the round's sampled entries into
            # the device staging table (mask-keyed gather — the sample set
            # is deterministic from the Philox key, so scan_rows computes
            # it on-device before the gradient kernel needs it). Fully
            # async; no whole-table copy, no host synchronization.
            # ASYNCAMD_SPILL_FULLCOPY=1 restores the round-1 full-copy
            # behavior for A/B.
            import os
            if os.environ.get("ASYNCAMD_SPILL_FULLCOPY") == "1":
                self._alpha_dev_buf.copy_(alpha, non_blocking=True)
            else:
                ops.spill_refresh(self._alpha_dev_buf, alpha, sh.y,
                                  self._spill_rows, self._spill_y,
                                  self._spill_cnt, self._spill_cap,
                                  seed=cfg.seed, round_k=round_key,
                                  row_start=sh.row_start,
                                  rate=cfg.batch_rate)
            alpha_dev = self._alpha_dev_buf
        else:
            alpha_dev = alpha
        if sh.is_sparse:
            g, idx, e, n = ops.saga_grad_csr(
                sh.indptr, sh.indices, sh.values, sh.y, w, alpha_dev,
                seed=cfg.seed, round_k=round_key, row_start=sh.row_start,
                rate=cfg.batch_rate, objective=cfg.objective)
        else:
            g, idx, e, n = ops.saga_grad_dense(
                sh.X, sh.y, w, alpha_dev, seed=cfg.seed, round_k=round_key,
                row_start=sh.row_start, rate=cfg.batch_rate,
                objective=cfg.objective)
        return g, idx, e, n

    def _commit_pending(self) -> None:
        """Apply the staged history scalars from the last ACCEPTED round
        (the reference merges ScalarMap only inside the tau test,
        SparkASAGAThread.scala:191,206-208 — a rejected round's scalars are
        dropped; staging-until-accept reproduces that exactly)."""
        idx, e = self._pending_idx, self._pending_e
        if idx is None or int(idx.numel()) == 0:
            return
        if self.alpha.device.type == "cpu" and self.is_cuda:
            # kernel scatter into the pinned master (device-writable):
            # async on the worker stream, ordered before the next round's
            # spill_refresh gather on the same stream
            ops.saga_commit_pinned(self.alpha, idx, e)
        else:
            ops.saga_commit(self.alpha, idx, e)


class _nullctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
