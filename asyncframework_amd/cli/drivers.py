"""The five algorithm drivers — CLI-compatible with the reference.

Positional args (13) exactly as the reference drivers parse them
(SparkASGDThread.scala:39-51): [path name] [file name] [num columns]
[num rows] [num partitions] [num iterations] [step size] [taw] [batch rate]
[bucket ratio] [printer freq] [coeff] [seed]; the MLlib baseline takes 8
(SparkSGDMLLIB.scala:30-37). Extra engine flags (--device, --dtype,
--objective, --sparse, --history-placement) come after the positionals.

``file name`` = 'synthetic' generates data of the given shape instead of
loading a LibSVM file (no network in this environment).

Multi-GPU: launch any 13-arg driver under torchrun (one rank per GPU) and
it routes through the dist engine — ``numPart`` stays the LOGICAL worker
count (the reference's partitions), spread M = numPart/world per rank;
rank 0 prints the unchanged stdout contract:

    torchrun --nproc-per-node 8 -m asyncframework_amd.cli asgd-thread \
        synthetic synthetic 784 8100000 32 16000 1.5e-3 20000000 0.01 \
        0.7 200 -1 42 --device cuda --dtype bf16
(--engine native under torchrun selects the C++ rank-0 server,
csrc/server_dist.cpp)."""

from __future__ import annotations

import argparse
import sys
from typing import List, Optional

from ..engine.config import EngineConfig
from ..utils import logfmt
from .. import run as runner

ARG_NAMES_13 = ["path name", "file name", "num columns", "num rows",
                "num partitions", "num iterations", "step size", "taw",
                "batch rate", "bucket ratio", "printer freq", "coeff",
                "seed"]
ARG_NAMES_8 = ["path name", "file name", "num columns", "num rows",
               "num partitions", "num iterations", "step size", "batch rate"]


def _engine_flags(p: argparse.ArgumentParser) -> None:
    p.add_argument("--device", default="cpu",
                   help="cpu | cuda | cuda:N (default cpu)")
    p.add_argument("--dtype", default="fp32",
                   choices=["fp32", "fp64", "bf16", "fp16"])
    p.add_argument("--objective", default="lsq", choices=["lsq", "logistic"])
    p.add_argument("--sparse", action="store_true",
                   help="CSR data path (rcv1-shape)")
    p.add_argument("--history-placement", default="device",
                   choices=["device", "host"],
                   help="SAGA history table in HBM (device) or pinned host "
                        "DRAM (host spill, BASELINE config 5)")
    p.add_argument("--max-wall-s", type=float, default=None)
    p.add_argument("--engine", default="threads",
                   choices=["threads", "native"],
                   help="native = C++ event-loop engine (GPU, async only)")
    p.add_argument("--worker-timeout-s", type=float, default=0.0,
                   help="declare a busy worker dead after this many seconds "
                        "(0 = off; the reference leaves lost workers busy "
                        "forever)")
    p.add_argument("--checkpoint-path", default="",
                   help="periodic optimizer-state snapshot file")
    p.add_argument("--checkpoint-every", type=int, default=0,
                   help="checkpoint every N applied updates (0 = off)")
    p.add_argument("--resume-from", default="",
                   help="restore optimizer state from a checkpoint file "
                        "before running")


def _parse13(argv: List[str], prog: str):
    p = argparse.ArgumentParser(prog=prog)
    for name in ["pathname", "fname", "d", "N", "numPart", "numIter",
                 "gamma", "taw", "b", "bucketRatio", "printerFreq", "coeff",
                 "seed"]:
        p.add_argument(name)
    _engine_flags(p)
    a = p.parse_args(argv)
    return a


def _parse8(argv: List[str], prog: str):
    p = argparse.ArgumentParser(prog=prog)
    for name in ["pathname", "fname", "d", "N", "numPart", "numIter",
                 "gamma", "b"]:
        p.add_argument(name)
    _engine_flags(p)
    return p.parse_args(argv)


def _cfg13(a, algo: str, sync: bool) -> EngineConfig:
    return EngineConfig(
        d=int(a.d), N=int(a.N), num_workers=int(a.numPart),
        num_iterations=int(a.numIter), gamma=float(a.gamma),
        taw=int(a.taw), batch_rate=float(a.b),
        bucket_ratio=float(a.bucketRatio), printer_freq=int(a.printerFreq),
        delay_coeff=float(a.coeff), seed=int(a.seed), algo=algo, sync=sync,
        objective=a.objective, dtype=a.dtype, device=a.device,
        history_placement=a.history_placement,
        worker_timeout_s=a.worker_timeout_s,
        checkpoint_path=a.checkpoint_path,
        checkpoint_every=a.checkpoint_every)


def _run(cfg: EngineConfig, a, app: str, names, vals) -> None:
    import os
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        _run_dist(cfg, a, app, names, vals, world)
        return
    logfmt.print_header(app, names, vals)
    sparse = a.sparse
    data = runner.load_dataset(cfg, a.pathname, a.fname, sparse=sparse,
                               device=a.device)
    if sparse:
        workers = runner.build_csr_workers(cfg, *data)
    else:
        workers = runner.build_dense_workers(cfg, *data)
    res, _srv = runner.run_engine(cfg, workers, max_wall_s=a.max_wall_s,
                                  engine=a.engine, resume_from=a.resume_from)
    runner.final_report(cfg, res, data, sparse, device=a.device)


def _run_dist(cfg: EngineConfig, a, app: str, names, vals,
              world: int) -> None:
    """torchrun path: one process per GPU; numPart logical workers spread
    M per rank (the reference's partitions-independent-of-executors model).
    Rank 0 hosts the server and prints the stdout contract; every rank
    loads the dataset once (rank 0 needs it whole for the objective-sweep
    epilogue) and keeps only its shards."""
    import os

    import torch
    import torch.distributed as dist

    from ..data.shard import row_shards
    from ..engine.worker import Shard, Worker

    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    device = torch.device(a.device)
    if device.type == "cuda":
        device = torch.device(f"cuda:{local_rank}")
        torch.cuda.set_device(device)
    assert cfg.num_workers % world == 0, \
        "numPart must be a multiple of the torchrun world size"
    M = cfg.num_workers // world
    cfg.device = str(device)
    if rank == 0:
        logfmt.print_header(app, names, vals)
    sparse = a.sparse
    data = runner.load_dataset(cfg, a.pathname, a.fname, sparse=sparse,
                               device="cpu")
    shards = row_shards(cfg.N, cfg.num_workers)
    workers = []
    for j in range(M):
        wid = rank * M + j
        s, t = shards[wid]
        if sparse:
            indptr, indices, values, y = data
            base = int(indptr[s])
            sh = Shard(row_start=s, n_rows=t - s,
                       indptr=(indptr[s:t + 1] - base).to(device),
                       indices=indices[base:int(indptr[t])].to(device),
                       values=values[base:int(indptr[t])].to(device),
                       y=y[s:t].to(device))
        else:
            X, y = data
            sh = Shard(row_start=s, n_rows=t - s, X=X[s:t].to(device),
                       y=y[s:t].to(device))
        workers.append(Worker(wid, sh, cfg, device=device))
    dist.init_process_group("nccl" if device.type == "cuda" else "gloo")
    if cfg.algo == "mllib":
        # the MLlib treeAggregate baseline maps to all_reduce:
        # replicated weights, the collective IS the barrier
        from ..engine.dist_sync import AllReduceSyncEngine
        eng = AllReduceSyncEngine(cfg, workers, device)
        res = eng.run(max_wall_s=a.max_wall_s, verbose=(rank == 0))
        if rank == 0:
            runner.final_report(cfg, res, data, sparse, device="cpu")
        dist.barrier()
        dist.destroy_process_group()
        return
    if a.engine == "native":
        assert not cfg.sync, "--engine native under torchrun is async-only"
        from ..engine.dist_native import NativeDistEngine
        eng = NativeDistEngine(cfg, workers, device)
        res = eng.run(max_wall_s=a.max_wall_s, verbose=(rank == 0),
                      resume_from=a.resume_from)
    else:
        from ..engine.dist import DistEngine
        eng = DistEngine(cfg, workers, device)
        res = eng.run(max_wall_s=a.max_wall_s, verbose=(rank == 0),
                      resume_from=a.resume_from)
    if rank == 0:
        runner.final_report(cfg, res, data, sparse, device="cpu")
    dist.destroy_process_group()


def _vals13(a):
    return [a.pathname, a.fname, a.d, a.N, a.numPart, a.numIter, a.gamma,
            a.taw, a.b, a.bucketRatio, a.printerFreq, a.coeff, a.seed]


def asgd_thread(argv: Optional[List[str]] = None) -> None:
    """Async bounded-staleness SGD (reference SparkASGDThread)."""
    a = _parse13(argv if argv is not None else sys.argv[1:], "asgd-thread")
    cfg = _cfg13(a, "asgd", sync=False)
    _run(cfg, a, "ASGD", ARG_NAMES_13, _vals13(a))


def asgd_sync(argv: Optional[List[str]] = None) -> None:
    """Synchronous SGD with user-space barrier (reference SparkASGDSync)."""
    a = _parse13(argv if argv is not None else sys.argv[1:], "asgd-sync")
    cfg = _cfg13(a, "asgd", sync=True)
    _run(cfg, a, "ASGDSync", ARG_NAMES_13, _vals13(a))


def asaga_thread(argv: Optional[List[str]] = None) -> None:
    """Async SAGA with per-sample gradient history (reference
    SparkASAGAThread)."""
    a = _parse13(argv if argv is not None else sys.argv[1:], "asaga-thread")
    cfg = _cfg13(a, "asaga", sync=False)
    _run(cfg, a, "ASAGA", ARG_NAMES_13, _vals13(a))


def asaga_sync(argv: Optional[List[str]] = None) -> None:
    """Synchronous SAGA (reference SparkASAGASync)."""
    a = _parse13(argv if argv is not None else sys.argv[1:], "asaga-sync")
    cfg = _cfg13(a, "asaga", sync=True)
    _run(cfg, a, "ASAGASync", ARG_NAMES_13, _vals13(a))


def sgd_mllib(argv: Optional[List[str]] = None) -> None:
    """MLlib mini-batch SGD baseline (reference SparkSGDMLLIB +
    GradientDescent.runMiniBatchSGD with the 100-iteration weight-history
    hook, GradientDescent.scala:255-260)."""
    a = _parse8(argv if argv is not None else sys.argv[1:], "sgd-mllib")
    cfg = EngineConfig(
        d=int(a.d), N=int(a.N), num_workers=int(a.numPart),
        num_iterations=int(a.numIter), gamma=float(a.gamma),
        batch_rate=float(a.b), algo="mllib", sync=True, printer_freq=100,
        objective=a.objective, dtype=a.dtype, device=a.device, seed=42,
        delay_coeff=0.0)
    import os
    vals = [a.pathname, a.fname, a.d, a.N, a.numPart, a.numIter, a.gamma,
            a.b]
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        a.resume_from = ""
        _run_dist(cfg, a, "MLlib SGD", ARG_NAMES_8, vals, world)
        return
    logfmt.print_header("MLlib SGD", ARG_NAMES_8, vals)
    data = runner.load_dataset(cfg, a.pathname, a.fname, sparse=a.sparse,
                               device=a.device)
    if a.sparse:
        workers = runner.build_csr_workers(cfg, *data)
    else:
        workers = runner.build_dense_workers(cfg, *data)
    res, _ = runner.run_engine(cfg, workers, max_wall_s=a.max_wall_s)
    runner.final_report(cfg, res, data, a.sparse, device=a.device)
