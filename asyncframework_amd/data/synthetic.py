"""Synthetic dataset generators matching the BASELINE shapes.

The reference's datasets (mnist8m 8.1Mx784 dense, epsilon 400kx2000 dense,
rcv1_full.binary 697kx47236 sparse; reference README.md) are not fetchable in
this environment — BASELINE.json mandates synthetic data of those shapes with
random-init weights. Generators are seeded and can emit directly on a GPU
device so a 12.7 GB mnist8m-shape tensor never round-trips through host RAM.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

SHAPES = {
    "mnist8m": (8_100_000, 784),
    "epsilon": (400_000, 2_000),
    "rcv1": (697_641, 47_236),
    "tiny": (1_000, 784),  # BASELINE config 1 plumbing shape
}


def synthetic_dense(n_rows: int, n_cols: int, *, seed: int = 42,
                    dtype: torch.dtype = torch.float32,
                    device: str | torch.device = "cpu",
                    objective: str = "lsq",
                    w_true: Optional[torch.Tensor] = None,
                    noise: float = 0.01,
                    chunk_rows: int = 1 << 20
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Dense (X, y) with a planted model so loss curves are meaningful.

    X ~ N(0, 1/sqrt(d)); lsq: y = X w* + noise; logistic: y = Bernoulli-free
    deterministic label sign(X w*) in {0,1} (keeps y reproducible across
    devices). Generated chunk-wise on the target device."""
    dev = torch.device(device)
    gen = torch.Generator(device=dev)
    gen.manual_seed(seed)
    scale = 1.0 / float(np.sqrt(n_cols))
    X = torch.empty((n_rows, n_cols), dtype=dtype, device=dev)
    y = torch.empty(n_rows, dtype=torch.float32, device=dev)
    if w_true is None:
        w_true = torch.randn(n_cols, generator=gen, device=dev,
                             dtype=torch.float32)
    for s in range(0, n_rows, chunk_rows):
        t = min(s + chunk_rows, n_rows)
        Xc = torch.randn((t - s, n_cols), generator=gen, device=dev,
                         dtype=torch.float32).mul_(scale)
        z = Xc @ w_true
        if objective == "logistic":
            y[s:t] = (z > 0).float()
        else:
            if noise > 0:
                z = z + noise * torch.randn(t - s, generator=gen, device=dev)
            y[s:t] = z
        X[s:t] = Xc.to(dtype)
    return X, y


def synthetic_csr(n_rows: int, n_cols: int, *, nnz_per_row: int = 73,
                  seed: int = 42, device: str | torch.device = "cpu",
                  objective: str = "lsq",
                  dtype: torch.dtype = torch.float32
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Sparse CSR (indptr, indices, values, y), rcv1-like (~73 nnz/row,
    reference README datasets). Column draw is uniform; values N(0,1)/sqrt(nnz).
    Built with numpy then moved to device."""
    rng = np.random.default_rng(seed)
    counts = np.maximum(1, rng.poisson(nnz_per_row, size=n_rows))
    indptr = np.zeros(n_rows + 1, dtype=np.int32)
    np.cumsum(counts, out=indptr[1:])
    nnz = int(indptr[-1])
    indices = rng.integers(0, n_cols, size=nnz, dtype=np.int64).astype(np.int32)
    values = (rng.standard_normal(nnz) / np.sqrt(nnz_per_row)).astype(np.float32)
    # row-wise sort of indices (vectorized via argsort on (row, col))
    rows = np.repeat(np.arange(n_rows, dtype=np.int64), counts)
    order = np.lexsort((indices, rows))
    indices = indices[order]
    values = values[order]
    w_true = rng.standard_normal(n_cols).astype(np.float32)
    # y via sparse matvec
    z = np.zeros(n_rows, dtype=np.float64)
    np.add.at(z, rows, values.astype(np.float64) * w_true[indices])
    if objective == "logistic":
        y = (z > 0).astype(np.float32)
    else:
        y = (z + 0.01 * rng.standard_normal(n_rows)).astype(np.float32)
    dev = torch.device(device)
    return (torch.from_numpy(indptr).to(dev),
            torch.from_numpy(indices).to(dev),
            torch.from_numpy(values).to(dev).to(dtype),
            torch.from_numpy(y).to(dev))
