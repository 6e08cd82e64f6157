"""High-level run API shared by the five CLI drivers and the bench.

Builds shards/workers/engine from an EngineConfig + dataset, runs it, and
emits the reference's stdout contract (SURVEY §2.1 "API surface to keep
compatible")."""

from __future__ import annotations

from typing import List, Optional

import torch

from . import ops
from .data.shard import row_shards
from .data.synthetic import synthetic_csr, synthetic_dense
from .data.libsvm import load_libsvm
from .engine.config import EngineConfig
from .engine.delay import DelayInjector
from .engine.local import AsyncEngine, RunResult, SyncEngine
from .engine.server import Server
from .engine.worker import Shard, Worker
from .utils import logfmt


def load_dataset(cfg: EngineConfig, pathname: str, fname: str,
                 sparse: bool = False, device: str = "cpu"):
    """'synthetic' fname -> generated data of (cfg.N, cfg.d); otherwise a
    LibSVM file at pathname+fname (reference MLUtils.loadLibSVMFile)."""
    dt = cfg.torch_dtype()
    if fname.startswith("synthetic") or pathname.startswith("synthetic"):
        if sparse:
            return synthetic_csr(cfg.N, cfg.d, seed=cfg.seed, device=device,
                                 objective=cfg.objective, dtype=dt)
        return synthetic_dense(cfg.N, cfg.d, seed=cfg.seed, device=device,
                               objective=cfg.objective, dtype=dt)
    path = pathname + fname
    if sparse:
        return load_libsvm(path, n_features=cfg.d, dense=False, dtype=dt,
                           device=device)
    return load_libsvm(path, n_features=cfg.d, dense=True, dtype=dt,
                       device=device)


def build_dense_workers(cfg: EngineConfig, X: torch.Tensor, y: torch.Tensor,
                        devices: Optional[List[torch.device]] = None
                        ) -> List[Worker]:
    """Shard rows over numPart workers (replaces the reference's
    repartition shuffle, SparkASGDThread.scala:76). ``devices`` maps worker
    -> device for multi-GPU single-process setups; default: data's device."""
    shards = row_shards(X.shape[0], cfg.num_workers)
    workers = []
    for wid, (s, t) in enumerate(shards):
        dev = devices[wid] if devices else X.device
        Xs = X[s:t].to(dev) if X.device != dev else X[s:t]
        ys = y[s:t].to(dev) if y.device != dev else y[s:t]
        workers.append(Worker(wid, Shard(row_start=s, n_rows=t - s, X=Xs,
                                         y=ys), cfg, device=dev))
    return workers


def build_csr_workers(cfg: EngineConfig, indptr, indices, values, y,
                      devices=None) -> List[Worker]:
    n_rows = indptr.shape[0] - 1
    shards = row_shards(n_rows, cfg.num_workers)
    workers = []
    for wid, (s, t) in enumerate(shards):
        dev = devices[wid] if devices else values.device
        base = int(indptr[s])
        ip = (indptr[s:t + 1] - base).to(dev)
        lo, hi = base, int(indptr[t])
        workers.append(Worker(wid, Shard(
            row_start=s, n_rows=t - s,
            indptr=ip, indices=indices[lo:hi].to(dev),
            values=values[lo:hi].to(dev), y=y[s:t].to(dev)),
            cfg, device=torch.device(dev)))
    return workers


def run_engine(cfg: EngineConfig, workers: List[Worker],
               max_wall_s: Optional[float] = None,
               verbose: bool = True, engine: str = "threads",
               resume_from: str = ""):
    """engine='threads' (the Python mailbox engine — CPU + semantics
    oracle) or 'native' (the C++ event loop, GPU only, async only).
    ``resume_from`` restores a checkpoint (threads engine only)."""
    if engine == "native":
        assert not resume_from, "resume is a threads-engine feature"
        assert not (cfg.checkpoint_every > 0 and cfg.checkpoint_path), \
            ("checkpointing with the single-GPU native engine is not wired "
             "— use --engine threads (the dist engines support it)")
        from .engine.native import NativeLocalEngine
        assert not cfg.sync, "native engine is async-only"
        neng = NativeLocalEngine(cfg, [w.shard for w in workers],
                                 workers[0].device)
        nres = neng.run(max_wall_s=max_wall_s or 1800.0,
                        snapshot_every=(cfg.printer_freq
                                        if cfg.snapshot_weights else 0))
        if verbose:
            for i in range(0, nres["k"], cfg.printer_freq):
                print(f"Iteration {i} is finished")
        waiting = {i: int(ms) for i, ms in enumerate(nres["waiting_ms"])}
        res = RunResult(k=int(nres["k"]), elapsed_ms=int(nres["elapsed_ms"]),
                        opt_vars=nres.get("opt_vars", []),
                        waiting_time=waiting, w=neng.w,
                        applied=int(nres["applied"]),
                        rejected=int(nres["rejected"]))
        return res, neng
    server = Server(cfg, device=workers[0].device)
    if resume_from:
        from .engine.checkpoint import load_checkpoint, restore
        restore(server, workers, load_checkpoint(resume_from))
    delay = DelayInjector(cfg.num_workers, cfg.delay_coeff, cfg.seed,
                          calib_window=cfg.calib_factor * cfg.num_workers)
    eng_cls = SyncEngine if cfg.sync else AsyncEngine
    eng = eng_cls(cfg, workers, server=server, delay=delay)
    eng.verbose = verbose
    res = eng.run(max_wall_s=max_wall_s)
    return res, server


def final_report(cfg: EngineConfig, res: RunResult, data, sparse: bool,
                 device: str = "cpu") -> None:
    """The shutdown epilogue: elapsed, waiting times, and the per-iterate
    objective sweep in one pass (reference SparkASGDThread.scala:346-410)."""
    logfmt.elapsed(res.elapsed_ms)
    logfmt.waiting_times(res.waiting_time, max(res.k, 1))
    if res.opt_vars:
        W = torch.stack([w for (_, w) in res.opt_vars]).to(device)
        if sparse:
            indptr, indices, values, y = data
            obj = ops.objective_sweep_csr(indptr, indices, values, y, W,
                                          cfg.objective, N_total=cfg.N)
        else:
            X, y = data
            obj = ops.objective_sweep(X, y, W, cfg.objective)
        logfmt.objective_lines(
            (t_ms, float(o)) for (t_ms, _), o in zip(res.opt_vars, obj))
    else:
        logfmt.objective_lines([])
