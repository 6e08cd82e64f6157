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
Checkpoint/resume is supported: a rank-0 monitor thread snapshots
(w, alpha_bar, k, clock, history tables — remote ones gathered through the
C++ snap sideband) with the same schema as the threads engine, so
checkpoints are interchangeable across engines."""

from __future__ import annotations

import importlib.util
import os
import threading
from typing import List, Optional

import torch
import torch.distributed as dist

from ..data.shard import row_shards
from .config import EngineConfig
from .dist import DistEngine, _send
from .local import RunResult
from .messages import HDR, Dispatch, pack_dispatch
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

    def _make_cfg(self, k0: int = 0, clock0: int = 0):
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
        c.k0 = k0
        c.clock0 = clock0
        c.bucket_ratio = cfg.bucket_ratio
        c.worker_timeout_s = cfg.worker_timeout_s
        from ..utils.trace import get_tracer
        c.trace = get_tracer() is not None
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

    # ---- checkpointing (same schema as engine/checkpoint.py) --------------
    def _gather_state(self):
        cfg, srv = self.cfg, self.srv
        state = {
            "k": srv.k(),
            "current_time": srv.clock(),
            "w": srv.weights().detach().cpu().clone(),
            "alpha_bar": (srv.alpha_bar().detach().cpu().clone()
                          if cfg.algo == "asaga" else None),
            "cfg": cfg.__dict__.copy(),
            "alpha": {},
        }
        for j, wk in enumerate(self.base.local_workers):
            if wk.alpha is not None:
                state["alpha"][j] = wk.alpha.detach().cpu().clone()
        if cfg.algo == "asaga":
            for wid in range(self.M, cfg.num_workers):
                srv.request_alpha_snapshot(wid)
            for wid in range(self.M, cfg.num_workers):
                if srv.wait_alpha(wid, 15.0):
                    t = srv.get_alpha(wid)
                    if t is not None:  # empty = peer already shut down
                        state["alpha"][wid] = t
        return state

    def _checkpoint_monitor(self, stop_ev: threading.Event):
        from .checkpoint import save_state
        cfg = self.cfg
        next_k = cfg.checkpoint_every
        while not stop_ev.is_set():
            k = self.srv.k()
            if k >= next_k and k < cfg.num_iterations:
                # (a snap queued after shutdown would never be answered —
                # the run's final state is still captured by the last
                # periodic snapshot, as in the threads engine)
                save_state(cfg.checkpoint_path, self._gather_state())
                next_k += cfg.checkpoint_every
            stop_ev.wait(0.05)

    def _push_alpha(self, wid: int, table: torch.Tensor) -> None:
        """Resume: send a restored history table to wid's rank BEFORE the
        C++ channel threads start (plain sends on the pair group are
        race-free then); the Python worker loop handles snap=2."""
        pg = self.base.pair_groups[wid]
        buf = torch.zeros(self.cfg.d + HDR, dtype=torch.float32,
                          device=self.device)
        pack_dispatch(buf, self.cfg.d, Dispatch(w=None, snap=2))
        peer = wid // self.M
        _send(buf, peer, pg)
        _send(table.to(dtype=torch.float32, device=self.device), peer, pg)

    def run(self, max_wall_s: Optional[float] = None,
            verbose: bool = True,
            resume_from: str = "") -> Optional[RunResult]:
        cfg = self.cfg
        if self.rank != 0:
            self.base.worker_loop()
            dist.barrier()
            return None
        core = _load()
        k0 = clock0 = 0
        w0 = torch.zeros(cfg.d, dtype=torch.float32, device=self.device)
        state = None
        if resume_from:
            from .checkpoint import load_checkpoint
            state = load_checkpoint(resume_from)
            ck_P = state.get("cfg", {}).get("num_workers")
            if ck_P is not None and ck_P != cfg.num_workers:
                raise ValueError(
                    f"checkpoint was taken with num_workers={ck_P}, cannot "
                    f"resume with num_workers={cfg.num_workers}")
            if state.get("cfg", {}).get("d") not in (None, cfg.d):
                raise ValueError("checkpoint dimensionality mismatch")
            k0 = int(state["k"])
            clock0 = int(state["current_time"])
            w0 = state["w"].to(self.device)
            for j, wk in enumerate(self.base.local_workers):
                if wk.alpha is not None and j in state["alpha"]:
                    wk.alpha.copy_(state["alpha"][j].to(wk.alpha.device))
        pgs = [self.base.pair_groups[wid]
               for wid in range(self.M, cfg.num_workers)]
        shards = row_shards(cfg.N, cfg.num_workers)
        alpha_rows = [(shards[wid][1] - shards[wid][0])
                      if cfg.algo == "asaga" else 0
                      for wid in range(self.M, cfg.num_workers)]
        self.srv = core.DistServer(self._make_cfg(k0, clock0), w0, pgs,
                                   [int(m) for m in self.mark_at],
                                   alpha_rows=alpha_rows)
        if state is not None:
            if cfg.algo == "asaga" and state.get("alpha_bar") is not None:
                self.srv.alpha_bar().copy_(
                    state["alpha_bar"].to(self.device))
            for wid in range(self.M, cfg.num_workers):
                if cfg.algo == "asaga" and wid in state["alpha"]:
                    self._push_alpha(wid, state["alpha"][wid])
        self.srv.start()
        threads = []
        for j, wk in enumerate(self.base.local_workers):
            th = threading.Thread(target=self._local_worker_loop,
                                  args=(j, wk), daemon=True,
                                  name=f"lworker-{j}")
            th.start()
            threads.append(th)
        ckpt_stop = None
        if cfg.checkpoint_every > 0 and cfg.checkpoint_path:
            ckpt_stop = threading.Event()
            ckpt_th = threading.Thread(
                target=self._checkpoint_monitor, args=(ckpt_stop,),
                daemon=True, name="ckpt-monitor")
            ckpt_th.start()
        self.srv.wait_done(max_wall_s or 1800.0)
        if ckpt_stop is not None:
            ckpt_stop.set()
            ckpt_th.join(timeout=30.0)
        self.srv.join()
        for th in threads:
            th.join(timeout=10.0)
        k = self.srv.k()
        from ..utils.trace import get_tracer, stop_trace
        tr = get_tracer()
        if tr is not None:
            names = {0: "dispatch", 1: "accept", 2: "reject"}
            for ts_s, wid, kind, kk, st in self.srv.trace_events():
                tr.instant(names[int(kind)], 0,
                           args={"k": int(kk), "wid": int(wid),
                                 "staleness": int(st)},
                           ts_us=tr.monotonic_s_to_us(ts_s))
            stop_trace()
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
