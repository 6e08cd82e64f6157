"""LibSVM-format loader -> dense or CSR torch tensors.

MI355X-native replacement for the reference's MLUtils.loadLibSVMFile
(reference mllib/src/main/scala/org/apache/spark/mllib/util/MLUtils.scala:
71-166): plain file parse into CSR arrays, optional densify, no Spark RDDs.
LibSVM lines are ``label idx:val idx:val ...`` with 1-based indices.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch


def _parse_native(path: str):
    """mmap single-pass C++ parser (csrc/libsvm_parser.cpp) when the native
    extension is built; returns numpy CSR arrays or None."""
    try:
        from .. import _hip_core
        return _hip_core.parse_libsvm(path)
    except (ImportError, AttributeError):
        return None


def load_libsvm(path: str, n_features: Optional[int] = None,
                dense: bool = False,
                dtype: torch.dtype = torch.float32,
                device: str | torch.device = "cpu"):
    """Returns (indptr, indices, values, y) CSR tensors, or (X, y) if
    ``dense``. Indices are converted to 0-based as in the reference loader
    (MLUtils.scala:91: ``indices.map(_ - 1)``). Uses the native C++ parser
    when available, with a pure-Python fallback."""
    nat = _parse_native(path)
    if nat is not None:
        indptr_a, cols_a, vals_a, y = nat
        d = (n_features if n_features is not None
             else (int(cols_a.max()) + 1 if cols_a.size else 0))
        dev = torch.device(device)
        if dense:
            n = len(y)
            X = np.zeros((n, d), dtype=np.float32)
            for r in range(n):
                s, t = indptr_a[r], indptr_a[r + 1]
                np.add.at(X[r], cols_a[s:t], vals_a[s:t])
            return (torch.from_numpy(X).to(dev).to(dtype),
                    torch.from_numpy(y).to(dev))
        return (torch.from_numpy(np.ascontiguousarray(indptr_a)).to(dev),
                torch.from_numpy(np.ascontiguousarray(cols_a)).to(dev),
                torch.from_numpy(np.ascontiguousarray(vals_a)).to(dev).to(dtype),
                torch.from_numpy(np.ascontiguousarray(y)).to(dev))
    labels = []
    indptr = [0]
    cols: list[int] = []
    vals: list[float] = []
    with open(path, "r") as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            parts = line.split()
            labels.append(float(parts[0]))
            for item in parts[1:]:
                i, v = item.split(":")
                cols.append(int(i) - 1)
                vals.append(float(v))
            indptr.append(len(cols))
    y = np.asarray(labels, dtype=np.float32)
    indptr_a = np.asarray(indptr, dtype=np.int32)
    cols_a = np.asarray(cols, dtype=np.int32)
    vals_a = np.asarray(vals, dtype=np.float32)
    d = n_features if n_features is not None else (int(cols_a.max()) + 1 if cols_a.size else 0)
    dev = torch.device(device)
    if dense:
        n = len(labels)
        X = np.zeros((n, d), dtype=np.float32)
        for r in range(n):
            s, t = indptr_a[r], indptr_a[r + 1]
            # accumulate duplicates (same as the native path's np.add.at)
            np.add.at(X[r], cols_a[s:t], vals_a[s:t])
        return torch.from_numpy(X).to(dev).to(dtype), torch.from_numpy(y).to(dev)
    return (torch.from_numpy(indptr_a).to(dev),
            torch.from_numpy(cols_a).to(dev),
            torch.from_numpy(vals_a).to(dev).to(dtype),
            torch.from_numpy(y).to(dev))
