"""In-process engine: worker threads + updater thread + dispatch loop.

Thread roles mirror the reference exactly:

* main thread      = the driver main loop (quorum gate, weight broadcast,
                     dispatch; SparkASGDThread.scala:230-345),
* updater thread   = the parameter server (mailbox drain, tau filter, weight
                     update, worker requeue; :153-226),
* worker threads   = executors running the fused gradient kernel on their
                     own HIP stream (Executor.TaskRunner analog).

On GPU all workers of one process share that process's device, each on its
own stream; the multi-process engine (engine.dist) wires remote ranks into
the same Server via channel proxies, so this file is the whole control plane.
"""

from __future__ import annotations

import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch

from .config import EngineConfig
from .delay import DelayInjector
from .messages import Dispatch, WorkerResult
from .server import Server
from .worker import Worker


@dataclass
class RunResult:
    k: int
    elapsed_ms: int
    opt_vars: List[Tuple[int, torch.Tensor]]
    waiting_time: Dict[int, int]
    w: torch.Tensor
    staleness_seen: List[int] = field(default_factory=list)
    applied: int = 0
    rejected: int = 0


class _LocalChannel:
    """Dispatch queue + thread for one in-process worker. Results flow
    straight into the server's completion path (the JobWaiter.taskSucceeded
    analog, reference JobWaiter.scala:56-60)."""

    spin_budget = 0  # >0: GIL-yielding hot-window poll after each round

    def __init__(self, worker: Worker, server: Server):
        self.worker = worker
        self.server = server
        self.q: "deque[Dispatch]" = deque()
        self._ev = threading.Event()
        self.thread = threading.Thread(target=self._loop, daemon=True,
                                       name=f"worker-{worker.id}")

    def start(self):
        self.thread.start()

    def dispatch(self, msg: Dispatch) -> None:
        self.q.append(msg)
        self._ev.set()

    def _loop(self):
        spin = 0
        while True:
            if not self.q:
                if spin > 0:  # hot window right after a round: the server
                    spin -= 1  # usually redispatches within ~100 us
                    time.sleep(0)
                    continue
                self._ev.wait(timeout=0.05)
                self._ev.clear()
                continue
            msg = self.q.popleft()
            if msg.stop:
                break
            res = self.worker.process(msg)
            self.server.on_completion(res)
            spin = self.spin_budget

    def join(self, timeout=None):
        self.thread.join(timeout)


class AsyncEngine:
    """Bounded-staleness asynchronous engine (ASGD/ASAGA)."""

    def __init__(self, cfg: EngineConfig, workers: Optional[List[Worker]] = None,
                 server: Optional[Server] = None,
                 delay: Optional[DelayInjector] = None,
                 channels: Optional[List] = None):
        self.cfg = cfg
        if channels is None:
            assert workers is not None and len(workers) == cfg.num_workers
            self.server = server or Server(cfg, device=workers[0].device)
            channels = [_LocalChannel(w, self.server) for w in workers]
        else:
            assert server is not None
            self.server = server
        self.delay = delay or DelayInjector(
            cfg.num_workers, cfg.delay_coeff, cfg.seed,
            calib_window=cfg.calib_factor * cfg.num_workers)
        self.channels = channels
        assert len(self.channels) == cfg.num_workers
        self.pending: "deque[int]" = deque(range(cfg.num_workers))
        self.dead: set = set()   # workers declared lost (worker_timeout_s)
        self._stop = threading.Event()
        self._pending_ev = threading.Event()
        self._pending_ev.set()
        # brief GIL-yielding spin before the blocking mailbox wait: wins
        # ~100 us/update when results arrive back-to-back (single-process
        # GPU engines), but CONTENDS the GIL against co-located worker and
        # proxy threads when rounds are long — dist mode disables it
        self.spin_budget = 100
        self.staleness_seen: List[int] = []
        self.accepted_staleness: List[int] = []
        self.applied = 0
        self.rejected = 0
        self.verbose = True
        # bench hooks: wall-clock stamps taken when k first reaches a mark
        self.mark_at = set()
        self.marks = {}

    def _local_workers(self):
        return [ch.worker for ch in self.channels if hasattr(ch, "worker")]

    def _gather_remote_alpha(self):
        """Pull SAGA history tables from remote ranks (dist engine snap
        sideband). Requests fan out first so peers snapshot concurrently;
        each table is a consistent point-in-time view of that worker's
        commit-on-accept history (cross-worker skew is inherent to an async
        checkpoint — same model as the in-process snapshot, which also
        observes workers mid-round)."""
        remotes = [(wid, ch) for wid, ch in enumerate(self.channels)
                   if hasattr(ch, "request_alpha") and ch.alpha_rows > 0
                   and wid not in self.dead]  # a dead peer never answers
        for _, ch in remotes:
            ch.request_alpha()
        out = {}
        for wid, ch in remotes:
            t = ch.wait_alpha(timeout=15.0)
            if t is not None:
                out[wid] = t
            else:
                import warnings
                warnings.warn(
                    f"worker {wid}: SAGA history snapshot timed out; the "
                    "checkpoint will have no alpha table for it (restore() "
                    "warns on resume)", RuntimeWarning, stacklevel=2)
        return out

    def _reap_dead_workers(self):
        """Failure detection the reference lacks (a lost task
        leaves a worker permanently busy): a worker whose round exceeds
        worker_timeout_s is declared dead and excluded from the quorum gate;
        if its result arrives later it is resurrected."""
        srv, cfg = self.server, self.cfg
        now = time.perf_counter()
        for wid in range(cfg.num_workers):
            if wid in self.dead:
                if srv.AC.STAT[wid].getAvailability():
                    self.dead.discard(wid)  # came back after all
                continue
            st = srv.AC.STAT[wid]
            sub = srv.submit_time.get(wid)
            if (not st.getAvailability() and sub is not None
                    and now - sub > cfg.worker_timeout_s):
                self.dead.add(wid)

    # -- server loop: mailbox drain + tau filter + update + requeue +
    #    quorum-gated dispatch. The reference splits this across its driver
    #    main loop and updater thread (SparkASGDThread.scala:153-226,
    #    230-345); folding them into ONE thread removes two cross-thread
    #    handoffs per update (measured: the threaded engine was handoff-
    #    bound at ~770 us/update with null workers).
    def _process_result(self, pr) -> None:
        cfg, srv = self.cfg, self.server
        res: WorkerResult = pr.gettaskResult()
        wid = pr.getWorkerID()
        now = time.perf_counter()
        if cfg.algo == "asaga":
            self.staleness_seen.append(srv.k - pr.getStaleness())
        else:
            self.staleness_seen.append(pr.getStaleness())
        from ..utils.trace import get_tracer
        tr = get_tracer()
        if self.accepts_now(pr):
            if tr is not None:
                tr.instant("accept", 0, args={"k": srv.k, "wid": wid,
                           "staleness": self.staleness_seen[-1]})
            self.accepted_staleness.append(self.staleness_seen[-1])
            srv.finish_time[wid] = now
            sub = srv.submit_time.get(wid)
            if sub is not None:
                self.delay.record_task(srv.k, (now - sub) * 1000.0)
            srv.apply(res)
            srv.last_accept[wid] = True
            self.pending.append(wid)
            if srv.k % cfg.printer_freq == 0:
                if self.verbose:
                    print(f"Iteration {srv.k} is finished")
                srv.maybe_log()
            srv.k += 1
            self.applied += 1
            if srv.k in self.mark_at:
                self.marks[srv.k] = time.perf_counter()
            if (cfg.checkpoint_every > 0 and cfg.checkpoint_path
                    and srv.k % cfg.checkpoint_every == 0):
                from .checkpoint import save_checkpoint
                save_checkpoint(cfg.checkpoint_path, srv,
                                self._local_workers(),
                                remote_alpha=self._gather_remote_alpha())
        else:
            if tr is not None:
                tr.instant("reject", 0, args={"k": srv.k, "wid": wid,
                           "staleness": self.staleness_seen[-1]})
            srv.last_accept[wid] = False
            self.pending.append(wid)
            self.rejected += 1

    def _dispatch_pending(self, first: bool = False) -> None:
        """Quorum gate + dispatch (reference main loop,
        SparkASGDThread.scala:230-345)."""
        cfg, srv = self.cfg, self.server
        if not self.pending:
            return
        alive = cfg.num_workers - len(self.dead)
        gate = min(cfg.gate, max(1, int(alive * cfg.bucket_ratio)))
        init_workers = (cfg.num_workers if first
                        else srv.available_workers())
        if init_workers < gate:
            return
        workers_list = []
        qsize = len(self.pending)
        for _ in range(qsize):
            workers_list.append(self.pending.popleft())
        from ..utils.trace import get_tracer
        tr = get_tracer()
        if tr is not None:
            tr.instant("dispatch", 0, args={"k": srv.k,
                                            "wids": list(workers_list)})
        self.delay.maybe_activate(srv.k)
        w_snap = srv.w.detach().clone()
        now = time.perf_counter()
        k_now = srv.k
        for wid in workers_list:
            prev_fin = srv.finish_time.get(wid, now)
            srv.waiting_time[wid] = (srv.waiting_time.get(wid, 0)
                                     + int((now - prev_fin) * 1000))
            srv.submit_time[wid] = now
            srv.AC.STAT[wid].setAvailability(False)
            msg = Dispatch(
                w=w_snap, ts=srv.AC.getCurrentTime(), k_submit=k_now,
                accept_prev=srv.last_accept.get(wid, True),
                delay_s=self.delay.delay_ms(wid, k_now) / 1000.0)
            self.channels[wid].dispatch(msg)

    def accepts_now(self, pr) -> bool:
        return self.server.accepts(pr)

    def run(self, max_wall_s: Optional[float] = None) -> RunResult:
        import queue as _q
        cfg, srv = self.cfg, self.server
        srv.start_time = time.perf_counter()
        if cfg.snapshot_weights and not srv.opt_vars:
            srv.opt_vars.append((0, srv.w.detach().cpu().clone()))
        for ch in self.channels:
            ch.start()
        t_start = time.perf_counter()
        self._dispatch_pending(first=True)
        while srv.k < cfg.num_iterations and not self._stop.is_set():
            if max_wall_s and time.perf_counter() - t_start > max_wall_s:
                self._stop.set()
                break
            # brief GIL-yielding spin before the blocking (Condition-based)
            # wait — saves a ~100 us thread wakeup per result when hot.
            # Peek the underlying deque directly: Queue.empty() takes the
            # mutex and contends with the producing worker threads.
            _dq = srv.AC.ResultList.queue
            spin = 0
            while not _dq and spin < self.spin_budget:
                time.sleep(0)
                spin += 1
            try:
                pr = srv.AC.ASYNCcollectAll(timeout=0.05)
            except _q.Empty:
                if cfg.worker_timeout_s > 0:
                    self._reap_dead_workers()
                    self._dispatch_pending()
                continue
            self._process_result(pr)
            while _dq and srv.k < cfg.num_iterations:
                self._process_result(srv.AC.ASYNCcollectAll())
            if cfg.worker_timeout_s > 0:
                self._reap_dead_workers()
            if srv.k < cfg.num_iterations:
                self._dispatch_pending()
        elapsed = srv.elapsed_ms()
        self._stop.set()
        for ch in self.channels:
            ch.dispatch(Dispatch(w=None, stop=True))
        for ch in self.channels:
            ch.join(timeout=10.0)
        from ..utils.trace import get_tracer, stop_trace
        if get_tracer() is not None:
            stop_trace()  # flush the event-log JSON (ASYNCAMD_TRACE)
        return RunResult(k=srv.k, elapsed_ms=elapsed, opt_vars=srv.opt_vars,
                         waiting_time=srv.waiting_time, w=srv.w,
                         staleness_seen=self.staleness_seen,
                         applied=self.applied, rejected=self.rejected)


class SyncEngine:
    """Synchronous variants: SparkASGDSync (user-space barrier counting all P
    results per round, average, step gamma/sqrt(k+1); reference
    SparkASGDSync.scala:239-277) and SparkASAGASync (full-barrier SAGA;
    SparkASAGASync.scala:263-304)."""

    def __init__(self, cfg: EngineConfig, workers: Optional[List[Worker]] = None,
                 server: Optional[Server] = None,
                 delay: Optional[DelayInjector] = None,
                 channels: Optional[List] = None):
        self.cfg = cfg
        if channels is None:
            assert workers is not None
            self.server = server or Server(cfg, device=workers[0].device)
            channels = [_LocalChannel(w, self.server) for w in workers]
        else:
            assert server is not None
            self.server = server
        self.delay = delay or DelayInjector(
            cfg.num_workers, cfg.delay_coeff, cfg.seed,
            calib_window=cfg.calib_factor * cfg.num_workers)
        self.channels = channels
        self.verbose = True
        self.mark_at = set()
        self.marks = {}

    def run(self, max_wall_s: Optional[float] = None) -> RunResult:
        import math
        cfg, srv = self.cfg, self.server
        srv.start_time = time.perf_counter()
        for ch in self.channels:
            ch.start()
        P = cfg.num_workers
        t_start = time.perf_counter()
        # resume-aware: a restored server continues from its k (fresh runs
        # start at 0); the step-size schedule gamma/sqrt(k+1) continues too
        for k in range(srv.k, cfg.num_iterations):
            if max_wall_s and time.perf_counter() - t_start > max_wall_s:
                break
            self.delay.maybe_activate(k)
            w_snap = srv.w.detach().clone()
            now = time.perf_counter()
            for wid in range(P):
                srv.submit_time[wid] = now
                msg = Dispatch(w=w_snap, ts=srv.AC.getCurrentTime(),
                               k_submit=k, accept_prev=True,
                               delay_s=self.delay.delay_ms(wid, k) / 1000.0)
                self.channels[wid].dispatch(msg)
            acc = torch.zeros_like(srv.w)
            got = 0
            nrows_round = 0
            while got < P:
                pr = srv.AC.ASYNCcollectAll(timeout=60.0)
                res: WorkerResult = pr.gettaskResult()
                wid = pr.getWorkerID()
                now2 = time.perf_counter()
                sub = srv.submit_time.get(wid)
                if sub is not None:
                    self.delay.record_task(k * P, (now2 - sub) * 1000.0)
                g = res.g
                if g.device != srv.device:
                    g = g.to(srv.device)
                acc += g
                nrows_round += res.nrows
                got += 1
            from .. import ops
            if cfg.algo == "asaga":
                # SparkASAGASync.scala:300-304: parRecs = b*N
                ops.saga_update(srv.w, acc, srv.alpha_bar, cfg.gamma,
                                1.0 / (cfg.batch_rate * cfg.N), 1.0 / cfg.N)
            elif cfg.algo == "mllib":
                # MLlib GradientDescent.runMiniBatchSGD semantics: the
                # gradient sum is divided by the ACTUAL minibatch size and
                # stepped with stepSize/sqrt(iter), iter from 1 (reference
                # mllib/.../optimization/GradientDescent.scala:287-290,
                # SimpleUpdater).
                gamma_k = cfg.gamma / math.sqrt(k + 1)
                ops.sgd_update(srv.w, acc, gamma_k,
                               1.0 / max(nrows_round, 1))
            else:
                # SparkASGDSync.scala:273-277
                gamma_k = cfg.gamma / math.sqrt(k + 1)
                ops.sgd_update(srv.w, acc, gamma_k,
                               1.0 / (cfg.batch_rate * cfg.N))
            if k % cfg.printer_freq == 0:
                if self.verbose:
                    print(f"Iteration {k} is finished")
                srv.k = k
                srv.maybe_log()
            srv.k = k + 1
            if srv.k in self.mark_at:
                self.marks[srv.k] = time.perf_counter()
            if (cfg.checkpoint_every > 0 and cfg.checkpoint_path
                    and srv.k % cfg.checkpoint_every == 0):
                from .checkpoint import save_checkpoint
                save_checkpoint(cfg.checkpoint_path, srv,
                                [ch.worker for ch in self.channels
                                 if hasattr(ch, "worker")])
        elapsed = srv.elapsed_ms()
        for ch in self.channels:
            ch.dispatch(Dispatch(w=None, stop=True))
        for ch in self.channels:
            ch.join(timeout=10.0)
        return RunResult(k=srv.k, elapsed_ms=elapsed, opt_vars=srv.opt_vars,
                         waiting_time=srv.waiting_time, w=srv.w)
