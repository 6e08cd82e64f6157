"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, plain torch on CPU.

On a GPU box the HIP extension (``asyncframework_amd._hip``, built in-tree by
``setup.py build_ext --inplace`` for gfx950) is REQUIRED: ops on CUDA tensors
raise if it is missing, so a silent eager fallback can never masquerade as
the native path. The torch reference path (ops.torch_ref) serves CPU tensors
and the numerics tests. Set ASYNCAMD_ALLOW_FALLBACK=1 to explicitly permit
torch fallback on GPU (debug only)."""

from __future__ import annotations

import os
from typing import Optional, Tuple

import numpy as np
import torch

from . import torch_ref
from ..utils.philox import bernoulli_mask

_hip = None
_hip_err: Optional[str] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        from . import hip as mod  # adapter over the in-tree _hip_core .so
        _hip = mod
    except ImportError as e:  # pragma: no cover - exercised on GPU box only
        _hip_err = str(e)
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _require_hip(op: str):
    mod = _load_hip()
    if mod is None:
        if os.environ.get("ASYNCAMD_ALLOW_FALLBACK") == "1":
            return None
        raise RuntimeError(
            f"{op}: HIP extension asyncframework_amd._hip_core is not built "
            f"(import error: {_hip_err}). On a GPU box the native kernels are "
            f"mandatory — run `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Set ASYNCAMD_ALLOW_FALLBACK=1 only "
            f"for debugging.")
    return mod


_OBJ_CODE = {"lsq": 0, "logistic": 1}


def make_mask(device: torch.device, seed: int, round_k: int, row_start: int,
              n_rows: int, rate: float) -> torch.Tensor:
    """Philox Bernoulli mask as a bool tensor (CPU path; GPU kernels compute
    the identical mask in-kernel from the same counters — csrc/philox.h)."""
    m = bernoulli_mask(seed, round_k, row_start, n_rows, rate)
    return torch.from_numpy(m).to(device)


def grad_dense(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor, *,
               seed: int, round_k: int, row_start: int, rate: float,
               objective: str = "lsq",
               out: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, int]:
    """Fused sample-mask + minibatch gradient (kernel K1 of SURVEY §2.5)."""
    if X.is_cuda:
        mod = _require_hip("grad_dense")
        if mod is not None:
            if out is None:
                out = torch.zeros(X.shape[1], dtype=torch.float32, device=X.device)
            else:
                out.zero_()
            n = mod.grad_dense(X, y, w, out, int(seed), int(round_k),
                               int(row_start), float(rate),
                               _OBJ_CODE[objective])
            return out, n
    mask = make_mask(X.device, seed, round_k, row_start, X.shape[0], rate)
    g, n = torch_ref.grad_dense(X, y, w, mask, objective)
    if out is not None:
        out.copy_(g)
        return out, n
    return g, n


def grad_csr(indptr: torch.Tensor, indices: torch.Tensor, values: torch.Tensor,
             y: torch.Tensor, w: torch.Tensor, *, seed: int, round_k: int,
             row_start: int, rate: float, objective: str = "lsq",
             out: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, int]:
    """Fused sample-mask + CSR minibatch gradient (kernel K2)."""
    if w.is_cuda:
        mod = _require_hip("grad_csr")
        if mod is not None:
            if out is None:
                out = torch.zeros(w.shape[0], dtype=torch.float32, device=w.device)
            else:
                out.zero_()
            n = mod.grad_csr(indptr, indices, values, y, w, out, int(seed),
                             int(round_k), int(row_start), float(rate),
                             _OBJ_CODE[objective])
            return out, n
    n_rows = indptr.shape[0] - 1
    mask = make_mask(w.device, seed, round_k, row_start, n_rows, rate)
    g, n = torch_ref.grad_csr(indptr, indices, values, y, w, mask, objective)
    if out is not None:
        out.copy_(g)
        return out, n
    return g, n


def saga_grad_dense(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
                    alpha: torch.Tensor, *, seed: int, round_k: int,
                    row_start: int, rate: float, objective: str = "lsq"
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """Fused mask + SAGA history gather + corrected gradient (kernel K3).
    Returns (g, sampled_idx, e_new, n); history commit is separate
    (saga_commit) so rejected rounds can be discarded."""
    if X.is_cuda:
        mod = _require_hip("saga_grad_dense")
        if mod is not None:
            g = torch.zeros(X.shape[1], dtype=torch.float32, device=X.device)
            idx, e = mod.saga_grad_dense(X, y, w, alpha, g, int(seed),
                                         int(round_k), int(row_start),
                                         float(rate), _OBJ_CODE[objective])
            return g, idx, e, int(idx.numel())
    mask = make_mask(X.device, seed, round_k, row_start, X.shape[0], rate)
    return torch_ref.saga_grad_dense(X, y, w, alpha, mask, objective)


def saga_grad_csr(indptr: torch.Tensor, indices: torch.Tensor,
                  values: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
                  alpha: torch.Tensor, *, seed: int, round_k: int,
                  row_start: int, rate: float, objective: str = "lsq"
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    if w.is_cuda:
        mod = _require_hip("saga_grad_csr")
        if mod is not None:
            g = torch.zeros(w.shape[0], dtype=torch.float32, device=w.device)
            idx, e = mod.saga_grad_csr(indptr, indices, values, y, w, alpha,
                                       g, int(seed), int(round_k),
                                       int(row_start), float(rate),
                                       _OBJ_CODE[objective])
            return g, idx, e, int(idx.numel())
    n_rows = indptr.shape[0] - 1
    mask = make_mask(w.device, seed, round_k, row_start, n_rows, rate)
    return torch_ref.saga_grad_csr(indptr, indices, values, y, w, alpha, mask,
                                   objective)


def saga_commit(alpha: torch.Tensor, idx: torch.Tensor, e: torch.Tensor) -> None:
    """Commit new history scalars alpha[idx] = e (the accepted-round analog of
    the reference's ScalarMap merge, SparkASAGAThread.scala:206-208)."""
    if alpha.is_cuda:
        mod = _require_hip("saga_commit")
        if mod is not None:
            mod.saga_commit(alpha, idx, e)
            return
    alpha[idx] = e.to(alpha.dtype)


def saga_commit_pinned(alpha_pinned: torch.Tensor, idx: torch.Tensor,
                       e: torch.Tensor) -> None:
    """Spill-path commit: kernel scatter into the pinned-host master table
    (GPU-only; the CPU engines keep their tables device/host-resident)."""
    mod = _require_hip("saga_commit_pinned")
    if mod is not None:
        mod.saga_commit_pinned(alpha_pinned, idx, e)
        return
    alpha_pinned[idx.cpu().long()] = e.cpu().to(alpha_pinned.dtype)


def spill_refresh(alpha_dev, alpha_pinned, y, rowlist, ylist, cnt, cap, *,
                  seed, round_k, row_start, rate) -> None:
    """Spill-path staging refresh (scan mask + gather sampled entries)."""
    mod = _require_hip("spill_refresh")
    if mod is not None:
        mod.spill_refresh(alpha_dev, alpha_pinned, y, rowlist, ylist, cnt,
                          cap, seed=seed, round_k=round_k,
                          row_start=row_start, rate=rate)
        return
    alpha_dev.copy_(alpha_pinned, non_blocking=True)  # debug fallback


def sgd_update(w: torch.Tensor, g: torch.Tensor, gamma_k: float,
               inv_batch: float) -> None:
    """Fused weight update (kernel K5)."""
    if w.is_cuda:
        mod = _require_hip("sgd_update")
        if mod is not None:
            mod.sgd_update(w, g, float(gamma_k), float(inv_batch))
            return
    torch_ref.sgd_update(w, g, gamma_k, inv_batch)


def saga_update(w: torch.Tensor, g: torch.Tensor, alpha_bar: torch.Tensor,
                gamma: float, inv_batch: float, inv_N: float) -> None:
    """Fused SAGA triple-axpy update (kernel K6)."""
    if w.is_cuda:
        mod = _require_hip("saga_update")
        if mod is not None:
            mod.saga_update(w, g, alpha_bar, float(gamma), float(inv_batch),
                            float(inv_N))
            return
    torch_ref.saga_update(w, g, alpha_bar, gamma, inv_batch, inv_N)


def objective_sweep(X: torch.Tensor, y: torch.Tensor, W: torch.Tensor,
                    objective: str = "lsq") -> torch.Tensor:
    """Objective of stacked iterates (K7). GEMM-shaped: goes through the
    library GEMM (hipBLASLt via torch.matmul) on GPU — a plain GEMM is
    library territory, only the fused hot ops are hand-written."""
    return torch_ref.objective_sweep(X, y, W, objective)


def objective_sweep_csr(indptr, indices, values, y, W, objective="lsq",
                        N_total=None) -> torch.Tensor:
    return torch_ref.objective_sweep_csr(indptr, indices, values, y, W,
                                         objective, N_total)
