"""Plain-PyTorch reference implementations of the compute hot path.

These are the numerics oracle for the HIP kernels (every HIP kernel's test
compares against these in fp32), the CPU execution path for the no-GPU test
tier, and the fallback is *never* silently used on a GPU box (see ops.__init__).

Semantics mirror the reference's JVM hot loops:

* per-sample least-squares gradient g_i = (x_i.w - y_i) x_i
  (reference examples/.../SparkASGDThread.scala:423-438 ``gradfun``),
* the logistic link variant g_i = (sigmoid(x_i.w) - y_i) x_i (the BASELINE
  north-star's hot path),
* SAGA history correction g_i - alpha_i x_i with scalar history alpha
  (reference SparkASAGAThread.scala:380-385),
* objective sweep sum((x.w - y)^2)/N (reference SparkASGDThread.scala:389-404).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def _residual(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
              objective: str) -> torch.Tensor:
    """Per-row scalar gradient coefficient e_i: x_i.w - y_i for 'lsq',
    sigmoid(x_i.w) - y_i for 'logistic'."""
    z = (X.to(w.dtype) @ w).float()
    if objective == "lsq":
        return z - y.float()
    elif objective == "logistic":
        return torch.sigmoid(z) - y.float()
    raise ValueError(f"unknown objective {objective!r}")


def grad_dense(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
               mask: torch.Tensor, objective: str = "lsq") -> Tuple[torch.Tensor, int]:
    """Summed minibatch gradient over the masked rows: g = X_S^T e_S.

    Returns (g fp32 [d], n_sampled). Mirrors gradfun + the per-partition axpy
    fold (reference RDD.scala:1103-1123 reducePartition, comOp = axpy)."""
    idx = mask.nonzero(as_tuple=True)[0]
    n = int(idx.numel())
    if n == 0:
        return torch.zeros(X.shape[1], dtype=torch.float32, device=X.device), 0
    Xs = X[idx]
    e = _residual(Xs, y[idx], w, objective)
    g = (Xs.float().t() @ e)
    return g, n


def grad_csr(indptr: torch.Tensor, indices: torch.Tensor, values: torch.Tensor,
             y: torch.Tensor, w: torch.Tensor, mask: torch.Tensor,
             objective: str = "lsq") -> Tuple[torch.Tensor, int]:
    """CSR variant of grad_dense (reference sparse dot/axpy,
    mllib/.../linalg/BLAS.scala:74-90,134-160). g accumulated dense fp32."""
    d = w.shape[0]
    g = torch.zeros(d, dtype=torch.float32, device=w.device)
    idx = mask.nonzero(as_tuple=True)[0]
    n = int(idx.numel())
    wf = w.float()
    for i in idx.tolist():
        s, t = int(indptr[i]), int(indptr[i + 1])
        cols = indices[s:t].long()
        vals = values[s:t].float()
        z = torch.dot(vals, wf[cols])
        if objective == "logistic":
            e = torch.sigmoid(z) - y[i].float()
        else:
            e = z - y[i].float()
        g.index_add_(0, cols, e * vals)
    return g, n


def saga_grad_dense(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
                    alpha: torch.Tensor, mask: torch.Tensor,
                    objective: str = "lsq"
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """SAGA corrected gradient over masked rows.

    Per sampled row i: e_i = link(x_i.w) - y_i; contribution (e_i - a_i) x_i
    (reference SparkASAGAThread.scala:380-385: ``g_i - alpha_i*x_i`` with the
    rank-1 reconstruction alpha_i*x_i of the historical gradient).

    Returns (g fp32 [d], sampled_idx int64 [n] — *shard-local* row indices,
    e fp32 [n] — the new history scalars, n). The caller commits
    ``alpha[sampled_idx] = e`` only when the server accepts the round
    (reference merges ScalarMap only inside the tau test,
    SparkASAGAThread.scala:191,206-208)."""
    idx = mask.nonzero(as_tuple=True)[0]
    n = int(idx.numel())
    d = X.shape[1]
    if n == 0:
        z = torch.zeros(d, dtype=torch.float32, device=X.device)
        return z, idx, torch.zeros(0, dtype=torch.float32, device=X.device), 0
    Xs = X[idx]
    e = _residual(Xs, y[idx], w, objective)
    corr = e - alpha[idx].float()
    g = Xs.float().t() @ corr
    return g, idx, e, n


def saga_grad_csr(indptr: torch.Tensor, indices: torch.Tensor,
                  values: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
                  alpha: torch.Tensor, mask: torch.Tensor,
                  objective: str = "lsq"
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """CSR SAGA gradient; see saga_grad_dense."""
    d = w.shape[0]
    g = torch.zeros(d, dtype=torch.float32, device=w.device)
    idx = mask.nonzero(as_tuple=True)[0]
    n = int(idx.numel())
    e_out = torch.zeros(n, dtype=torch.float32, device=w.device)
    wf = w.float()
    for j, i in enumerate(idx.tolist()):
        s, t = int(indptr[i]), int(indptr[i + 1])
        cols = indices[s:t].long()
        vals = values[s:t].float()
        z = torch.dot(vals, wf[cols])
        if objective == "logistic":
            e = torch.sigmoid(z) - y[i].float()
        else:
            e = z - y[i].float()
        e_out[j] = e
        g.index_add_(0, cols, (e - alpha[i].float()) * vals)
    return g, idx, e_out, n


def objective_sweep(X: torch.Tensor, y: torch.Tensor, W: torch.Tensor,
                    objective: str = "lsq", batch_rows: int = 262144
                    ) -> torch.Tensor:
    """Objective of each of T logged iterates in one dataset pass.

    W is [T, d] (stacked iterates — the reference computes all logged
    objectives in a single job, SparkASGDThread.scala:389-398). Returns [T]
    fp64. lsq: sum((X w_t - y)^2)/N; logistic: mean log-loss."""
    T = W.shape[0]
    N = X.shape[0]
    out = torch.zeros(T, dtype=torch.float64, device=X.device)
    # bf16 X: cast the SMALL operand (W, [T,d]) once and run a bf16 MFMA
    # GEMM instead of converting every X chunk up to fp32 (the conversion
    # kernel costs as much as the GEMM itself — profiles/r02 MFMA capture)
    Wt = W.to(X.device, X.dtype if X.dtype == torch.bfloat16 else W.dtype)
    for s in range(0, N, batch_rows):
        Xb = X[s:s + batch_rows]
        yb = y[s:s + batch_rows].float()
        Z = (Xb @ Wt.t()).float()  # [B, T] — library GEMM (MFMA on gfx950)
        if objective == "lsq":
            out += ((Z - yb[:, None]) ** 2).double().sum(dim=0)
        else:
            zy = Z * (2.0 * yb[:, None] - 1.0)
            out += torch.nn.functional.softplus(-zy).double().sum(dim=0)
    return out / N


def objective_sweep_csr(indptr: torch.Tensor, indices: torch.Tensor,
                        values: torch.Tensor, y: torch.Tensor,
                        W: torch.Tensor, objective: str = "lsq",
                        N_total: Optional[int] = None) -> torch.Tensor:
    """CSR objective sweep via torch.sparse mm."""
    n_rows = indptr.shape[0] - 1
    N = N_total if N_total is not None else n_rows
    sp = torch.sparse_csr_tensor(indptr, indices, values.float(),
                                 size=(n_rows, W.shape[1]))
    Z = torch.sparse.mm(sp, W.float().t())
    yb = y.float()
    if objective == "lsq":
        out = ((Z - yb[:, None]) ** 2).double().sum(dim=0)
    else:
        zy = Z * (2.0 * yb[:, None] - 1.0)
        out = torch.nn.functional.softplus(-zy).double().sum(dim=0)
    return out / N


def sgd_update(w: torch.Tensor, g: torch.Tensor, gamma_k: float,
               inv_batch: float) -> None:
    """Fused scale+axpy weight update, in place:
    w -= gamma_k * (g * inv_batch). Reference updater thread
    SparkASGDThread.scala:188-192 (scalOp + axpyOp)."""
    w.add_(g.to(w.dtype), alpha=-(gamma_k * inv_batch))


def saga_update(w: torch.Tensor, g: torch.Tensor, alpha_bar: torch.Tensor,
                gamma: float, inv_batch: float, inv_N: float) -> None:
    """Fused SAGA triple-axpy, in place (reference
    SparkASAGAThread.scala:217-220):
    w -= gamma * (g*inv_batch); w -= gamma*alpha_bar; alpha_bar += g*inv_N."""
    gf = g.float()
    w.add_(gf.to(w.dtype), alpha=-(gamma * inv_batch))
    w.add_(alpha_bar.to(w.dtype), alpha=-gamma)
    alpha_bar.add_(gf, alpha=inv_N)
