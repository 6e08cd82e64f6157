"""Native multi-GPU engine: the rank-0 control plane in C++ (csrc/
server_dist.cpp) over torch.distributed pair groups — RCCL point-to-point
over xGMI on GPU, gloo on CPU (which is how the CPU test tier runs the
exact same C++ loop).

Topology and wire format are IDENTICAL to engine/dist.py (M logical workers
per rank, per-worker pair communicators, [d+8]-float messages): worker ranks
run the unchanged, validated Python ``remote_worker_loop``; only rank 0's
proxy threads + server loop move to C++ (the Python engine spends
~100-250 us of GIL-bound work per update there — ROADMAP item 1).

Status: CPU(gloo)-validated this round; first multi-GPU RCCL execution is
round 2 (the Python dist engine remains the default N>1 path until then).
Checkpoint/resume stays on the Python dist engine."""

from __future__ import annotations

import importlib.util
import os
import threading
from typing import List, Optional

import torch
import torch.distributed as dist

from .config import EngineConfig
from .dist import DistEngine
from .local import RunResult
from .messages import Dispatch
from .worker import Worker

_mod = None


def _load():
    """Import the prebuilt in-tree _dist_core.so (built by build_hip.py /
    __graft_entry__.build; torch must be imported first for symbols)."""
    global _mod
    if _mod is not None:
        return _mod
    here = os.path.dirname(os.path.abspath(__file__))
    so = os.path.join(here, "..", "_dist_build", "_dist_core.so")
    if not os.path.exists(so):
        raise RuntimeError(
            "_dist_core.so not built — run `python build_hip.py` "
            f"(expected at {so})")
    spec = importlib.util.spec_from_file_location("_dist_core", so)
    _mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(_mod)
    return _mod


def dist_core_available() -> bool:
    try:
        _load()
        return True
    except (RuntimeError, ImportError):
        return False


class NativeDistEngine:
    """Same call shape as DistEngine; rank 0 drives the C++ server."""

    def __init__(self, cfg: EngineConfig, local_workers, device: torch.device,
                 mark_at: Optional[List[int]] = None):
        assert not cfg.sync, "native dist server is async-only"
        # reuse DistEngine's group construction (same deterministic order +
        # eager communicator warm-up on every rank)
        self.base = DistEngine(cfg, local_workers, device)
        self.cfg = cfg
        self.device = device
        self.rank = self.base.rank
        self.M = self.base.M
        self.mark_at = mark_at or []
        self.marks = {}
        self.srv = None

    def _make_cfg(self):
        core = _load()
        c = core.DSCfg()
        cfg = self.cfg
        c.d = cfg.d
        c.P = cfg.num_workers
        c.M = self.M
        c.N = cfg.N
        c.num_iter = cfg.num_iterations
        c.printer_freq = cfg.printer_freq
        c.gamma = cfg.gamma
        c.batch_rate = cfg.batch_rate
        c.taw = cfg.taw
        c.gate = cfg.gate
        c.coeff = cfg.delay_coeff
        c.seed = cfg.seed
        c.calib_window = cfg.calib_factor * cfg.num_workers
        c.asaga = cfg.algo == "asaga"
        c.snapshot_weights = cfg.snapshot_weights
        return c

    def _local_worker_loop(self, wid: int, worker: Worker):
        srv = self.srv
        while True:
            w, ts, k_submit, accept, delay_s, stop = \
                srv.local_next_dispatch(wid)
            if stop:
                break
            res = worker.process(Dispatch(w=w, ts=ts, k_submit=k_submit,
                                          accept_prev=accept,
                                          delay_s=delay_s))
            g = res.g
            if g.dtype != torch.float32:
                g = g.float()
            srv.local_deliver(wid, g.contiguous(), res.ts, res.k_submit,
                              res.elapsed_ms)

    def run(self, max_wall_s: Optional[float] = None,
            verbose: bool = True) -> Optional[RunResult]:
        cfg = self.cfg
        if self.rank != 0:
            self.base.worker_loop()
            dist.barrier()
            return None
        core = _load()
        w0 = torch.zeros(cfg.d, dtype=torch.float32, device=self.device)
        pgs = [self.base.pair_groups[wid]
               for wid in range(self.M, cfg.num_workers)]
        self.srv = core.DistServer(self._make_cfg(), w0, pgs,
                                   [int(m) for m in self.mark_at])
        self.srv.start()
        threads = []
        for j, wk in enumerate(self.base.local_workers):
            th = threading.Thread(target=self._local_worker_loop,
                                  args=(j, wk), daemon=True,
                                  name=f"lworker-{j}")
            th.start()
            threads.append(th)
        self.srv.wait_done(max_wall_s or 1800.0)
        self.srv.join()
        for th in threads:
            th.join(timeout=10.0)
        k = self.srv.k()
        if verbose:
            for i in range(0, k, cfg.printer_freq):
                print(f"Iteration {i} is finished")
        self.marks = dict(self.srv.marks())
        waiting = {i: int(ms)
                   for i, ms in enumerate(self.srv.waiting_ms())}
        opt_vars = list(zip(self.srv.opt_ms(), self.srv.opt_w()))
        res = RunResult(k=k, elapsed_ms=self.srv.elapsed_ms(),
                        opt_vars=opt_vars, waiting_time=waiting,
                        w=self.srv.weights(),
                        applied=self.srv.applied(),
                        rejected=self.srv.rejected())
        dist.barrier()
        return res
