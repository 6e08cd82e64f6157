"""Numerics of the reference (torch) ops vs closed form — the analog of the
reference's GradientDescentSuite closed-form gradient check
(mllib/src/test/.../optimization/GradientDescentSuite.scala:107-143)."""

import numpy as np
import pytest
import torch

from asyncframework_amd import ops
from asyncframework_amd.ops import torch_ref
from asyncframework_amd.data.synthetic import synthetic_csr, synthetic_dense


def _dense_case(n=64, d=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(n, d, generator=g)
    y = torch.randn(n, generator=g)
    w = torch.randn(d, generator=g)
    return X, y, w


def test_lsq_grad_closed_form():
    X, y, w = _dense_case()
    mask = torch.ones(64, dtype=torch.bool)
    g, n = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert n == 64
    expected = X.t() @ (X @ w - y)
    assert torch.allclose(g, expected, atol=1e-5)


def test_logistic_grad_closed_form():
    X, y, w = _dense_case()
    mask = torch.ones(64, dtype=torch.bool)
    g, _ = torch_ref.grad_dense(X, y, w, mask, "logistic")
    expected = X.t() @ (torch.sigmoid(X @ w) - y)
    assert torch.allclose(g, expected, atol=1e-5)


def test_masked_grad_subset():
    X, y, w = _dense_case()
    mask = torch.zeros(64, dtype=torch.bool)
    mask[5] = mask[17] = True
    g, n = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert n == 2
    e5 = (X[5] @ w - y[5])
    e17 = (X[17] @ w - y[17])
    assert torch.allclose(g, e5 * X[5] + e17 * X[17], atol=1e-5)


def test_empty_mask():
    X, y, w = _dense_case()
    mask = torch.zeros(64, dtype=torch.bool)
    g, n = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert n == 0 and torch.all(g == 0)


def test_csr_matches_dense():
    indptr, indices, values, y = synthetic_csr(50, 40, nnz_per_row=8, seed=3)
    d = 40
    X = torch.zeros(50, d)
    for r in range(50):
        s, t = int(indptr[r]), int(indptr[r + 1])
        X[r].index_add_(0, indices[s:t].long(), values[s:t])
    w = torch.randn(d, generator=torch.Generator().manual_seed(1))
    mask = torch.rand(50, generator=torch.Generator().manual_seed(2)) < 0.5
    gd, nd = torch_ref.grad_dense(X, y, w, mask, "lsq")
    gs, ns = torch_ref.grad_csr(indptr, indices, values, y, w, mask, "lsq")
    assert nd == ns
    assert torch.allclose(gd, gs, atol=1e-4)


def test_saga_grad_and_commit():
    X, y, w = _dense_case(32, 8, seed=5)
    alpha = torch.randn(32, generator=torch.Generator().manual_seed(7))
    mask = torch.zeros(32, dtype=torch.bool)
    mask[[2, 9, 30]] = True
    g, idx, e, n = torch_ref.saga_grad_dense(X, y, w, alpha, mask, "lsq")
    assert n == 3
    assert idx.tolist() == [2, 9, 30]
    expected = torch.zeros(8)
    for i in [2, 9, 30]:
        ei = X[i] @ w - y[i]
        expected += (ei - alpha[i]) * X[i]
        assert torch.allclose(e[idx.tolist().index(i)], ei, atol=1e-5)
    assert torch.allclose(g, expected, atol=1e-5)
    ops.saga_commit(alpha, idx, e)
    for j, i in enumerate([2, 9, 30]):
        assert alpha[i] == e[j]


def test_saga_csr_matches_dense():
    indptr, indices, values, y = synthetic_csr(40, 30, nnz_per_row=6, seed=9)
    X = torch.zeros(40, 30)
    for r in range(40):
        s, t = int(indptr[r]), int(indptr[r + 1])
        X[r].index_add_(0, indices[s:t].long(), values[s:t])
    w = torch.randn(30, generator=torch.Generator().manual_seed(4))
    alpha = torch.randn(40, generator=torch.Generator().manual_seed(5))
    mask = torch.rand(40, generator=torch.Generator().manual_seed(6)) < 0.4
    gd, idxd, ed, nd = torch_ref.saga_grad_dense(X, y, w, alpha, mask, "lsq")
    gs, idxs, es, ns = torch_ref.saga_grad_csr(indptr, indices, values, y, w,
                                               alpha, mask, "lsq")
    assert nd == ns
    assert torch.equal(idxd, idxs)
    assert torch.allclose(ed, es, atol=1e-4)
    assert torch.allclose(gd, gs, atol=1e-4)


def test_objective_sweep_matches_manual():
    X, y, w = _dense_case(100, 10, seed=11)
    W = torch.stack([w, 2 * w, torch.zeros_like(w)])
    obj = torch_ref.objective_sweep(X, y, W, "lsq")
    for t in range(3):
        manual = ((X @ W[t] - y) ** 2).double().sum() / 100
        assert abs(float(obj[t]) - float(manual)) < 1e-4 * (1 + float(manual))


def test_objective_sweep_csr():
    indptr, indices, values, y = synthetic_csr(60, 20, nnz_per_row=5, seed=13)
    X = torch.zeros(60, 20)
    for r in range(60):
        s, t = int(indptr[r]), int(indptr[r + 1])
        X[r].index_add_(0, indices[s:t].long(), values[s:t])
    W = torch.randn(2, 20, generator=torch.Generator().manual_seed(3))
    od = torch_ref.objective_sweep(X, y, W, "lsq")
    os_ = torch_ref.objective_sweep_csr(indptr, indices, values, y, W, "lsq")
    assert torch.allclose(od, os_, atol=1e-6)


def test_sgd_update_inplace():
    w = torch.ones(4)
    g = torch.full((4,), 2.0)
    torch_ref.sgd_update(w, g, gamma_k=0.5, inv_batch=0.25)
    # w -= 0.5 * (2 * 0.25) = 0.25
    assert torch.allclose(w, torch.full((4,), 0.75))


def test_saga_update_inplace():
    w = torch.zeros(3)
    g = torch.ones(3)
    ab = torch.full((3,), 0.5)
    torch_ref.saga_update(w, g, ab, gamma=1.0, inv_batch=0.5, inv_N=0.1)
    # w = 0 - 1*(1*0.5) - 1*0.5 = -1.0 ; ab = 0.5 + 0.1
    assert torch.allclose(w, torch.full((3,), -1.0))
    assert torch.allclose(ab, torch.full((3,), 0.6))


def test_dispatch_layer_cpu_uses_philox_mask():
    X, y, w = _dense_case(128, 8, seed=21)
    g1, n1 = ops.grad_dense(X, y, w, seed=42, round_k=3, row_start=0,
                            rate=0.5, objective="lsq")
    from asyncframework_amd.utils.philox import bernoulli_mask
    mask = torch.from_numpy(bernoulli_mask(42, 3, 0, 128, 0.5))
    g2, n2 = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert n1 == n2
    assert torch.allclose(g1, g2)


def test_bf16_data_fp32_grad():
    X, y, w = _dense_case(64, 16, seed=23)
    Xb = X.bfloat16()
    mask = torch.ones(64, dtype=torch.bool)
    g, _ = torch_ref.grad_dense(Xb, y, w.bfloat16(), mask, "lsq")
    gf, _ = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert g.dtype == torch.float32
    rel = (g - gf).norm() / gf.norm()
    assert rel < 0.05
