"""GraphEngine (hipGraph device-resident round loop) equivalence tests:
the captured graph must produce the SAME iterates as (a) a plain-PyTorch
sequential reference and (b) the threaded mailbox engine with one worker."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from asyncframework_amd import run as runner
from asyncframework_amd.data.synthetic import synthetic_dense, synthetic_csr
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.graph import GraphEngine
from asyncframework_amd.engine.worker import Shard, Worker
from asyncframework_amd.ops import torch_ref
from asyncframework_amd.utils.philox import bernoulli_mask


def _cfg(**kw):
    base = dict(d=64, N=20_000, num_workers=1, num_iterations=64,
                gamma=0.3, taw=1 << 30, batch_rate=0.05, bucket_ratio=0.5,
                printer_freq=1 << 30, delay_coeff=0.0, seed=42,
                device="cuda:0", snapshot_weights=False)
    base.update(kw)
    return EngineConfig(**base)


def _seq_ref_asgd(cfg, X, y):
    w = torch.zeros(cfg.d, device=X.device)
    for k in range(cfg.num_iterations):
        mask = torch.from_numpy(
            bernoulli_mask(cfg.seed, k + 1, 0, cfg.N, cfg.batch_rate)
        ).to(X.device)
        g, _ = torch_ref.grad_dense(X.float(), y, w, mask, cfg.objective)
        gamma_k = cfg.gamma / math.sqrt(k // cfg.num_workers + 1)
        w -= gamma_k * g / cfg.par_recs
    return w


def test_graph_asgd_matches_sequential_ref():
    cfg = _cfg()
    X, y = synthetic_dense(cfg.N, cfg.d, seed=1, device="cuda:0")
    eng = GraphEngine(cfg, Shard(row_start=0, n_rows=cfg.N, X=X, y=y),
                      torch.device("cuda:0"), unroll=10)
    eng.run(cfg.num_iterations)
    assert eng.k == cfg.num_iterations
    w_ref = _seq_ref_asgd(cfg, X, y)
    rel = float((eng.w - w_ref).norm() / (w_ref.norm() + 1e-12))
    assert rel < 1e-4, rel


def test_graph_asgd_matches_threaded_engine():
    cfg = _cfg(num_iterations=40)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=2, device="cuda:0")
    eng = GraphEngine(cfg, Shard(row_start=0, n_rows=cfg.N, X=X, y=y),
                      torch.device("cuda:0"), unroll=7)
    eng.run(cfg.num_iterations)
    workers = runner.build_dense_workers(cfg, X, y)
    res, _ = runner.run_engine(cfg, workers, verbose=False)
    rel = float((eng.w - res.w).norm() / (res.w.norm() + 1e-12))
    assert rel < 1e-4, rel


def test_graph_asaga_matches_sequential_ref():
    cfg = _cfg(algo="asaga", gamma=0.05, num_iterations=50)
    X, y = synthetic_dense(cfg.N, cfg.d, seed=3, device="cuda:0")
    eng = GraphEngine(cfg, Shard(row_start=0, n_rows=cfg.N, X=X, y=y),
                      torch.device("cuda:0"), unroll=9)
    eng.run(cfg.num_iterations)
    # sequential reference with per-round accepted commits
    w = torch.zeros(cfg.d, device="cuda:0")
    ab = torch.zeros(cfg.d, device="cuda:0")
    alpha = torch.zeros(cfg.N, device="cuda:0")
    for k in range(cfg.num_iterations):
        mask = torch.from_numpy(
            bernoulli_mask(cfg.seed, k + 1, 0, cfg.N, cfg.batch_rate)).cuda()
        g, idx, e, _ = torch_ref.saga_grad_dense(X.float(), y, w, alpha,
                                                 mask, cfg.objective)
        torch_ref.saga_update(w, g, ab, cfg.gamma, 1.0 / cfg.par_recs,
                              1.0 / cfg.N)
        alpha[idx] = e
    rel = float((eng.w - w).norm() / (w.norm() + 1e-12))
    assert rel < 1e-4, rel
    rel_a = float((eng.alpha - alpha).norm() / (alpha.norm() + 1e-12))
    assert rel_a < 1e-4, rel_a


def test_graph_csr_runs_and_decreases():
    cfg = _cfg(d=512, N=10_000, gamma=1.0, num_iterations=60)
    indptr, indices, values, y = synthetic_csr(cfg.N, cfg.d, nnz_per_row=30,
                                               seed=4, device="cuda:0")
    sh = Shard(row_start=0, n_rows=cfg.N, indptr=indptr, indices=indices,
               values=values, y=y)
    eng = GraphEngine(cfg, sh, torch.device("cuda:0"))
    eng.run(cfg.num_iterations)
    obj = torch_ref.objective_sweep_csr(indptr, indices, values, y,
                                        torch.stack([torch.zeros(cfg.d,
                                                                 device="cuda:0"),
                                                     eng.w]), "lsq")
    assert float(obj[1]) < float(obj[0])


def test_graph_overlap_odd_tail_parity():
    """The overlap graph is captured for buffer parity cur==0 with an even
    unroll; odd-length step_rounds calls must re-align via tail rounds and
    still match the sequential reference."""
    cfg = _cfg(num_iterations=57)  # not a multiple of unroll, odd tails
    X, y = synthetic_dense(cfg.N, cfg.d, seed=11, device="cuda:0")
    eng = GraphEngine(cfg, Shard(row_start=0, n_rows=cfg.N, X=X, y=y),
                      torch.device("cuda:0"), unroll=10)
    eng.step_rounds(3)      # odd: forces tail rounds + parity realign
    eng.step_rounds(24)
    eng.step_rounds(30)
    torch.cuda.synchronize()
    assert eng.k == 57
    w_ref = _seq_ref_asgd(cfg, X, y)
    rel = float((eng.w - w_ref).norm() / (w_ref.norm() + 1e-12))
    assert rel < 1e-4, rel
