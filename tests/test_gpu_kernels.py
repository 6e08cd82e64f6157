"""HIP kernel numerics vs the plain-PyTorch fp32 reference, on MI355X.
Every kernel (K1/K2/K3/K5/K6 + the in-kernel Philox K8) is compared against
ops.torch_ref on the same philox mask."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from asyncframework_amd import ops
from asyncframework_amd.ops import torch_ref
from asyncframework_amd.utils.philox import bernoulli_mask
from asyncframework_amd.data.synthetic import synthetic_csr


def _require_hip():
    assert ops.hip_available(), "HIP extension must be built on GPU boxes"


def _dense(n, d, seed=0, dtype=torch.float32):
    g = torch.Generator(device="cuda").manual_seed(seed)
    X = torch.randn(n, d, generator=g, device="cuda", dtype=torch.float32)
    y = torch.randn(n, generator=g, device="cuda")
    w = torch.randn(d, generator=g, device="cuda")
    return X.to(dtype), y, w


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("objective", ["lsq", "logistic"])
def test_grad_dense_matches_ref(dtype, objective):
    _require_hip()
    n, d = 4096, 784
    X, y, w = _dense(n, d, seed=1, dtype=dtype)
    g, cnt = ops.grad_dense(X, y, w, seed=42, round_k=7, row_start=1000,
                            rate=0.3, objective=objective)
    mask = torch.from_numpy(bernoulli_mask(42, 7, 1000, n, 0.3)).cuda()
    g_ref, cnt_ref = torch_ref.grad_dense(X.float(), y, w, mask, objective)
    assert cnt == cnt_ref
    tol = 2e-2 if dtype == torch.bfloat16 else 2e-4
    rel = float((g - g_ref).norm() / (g_ref.norm() + 1e-12))
    assert rel < tol, f"rel err {rel}"


def test_grad_dense_nondivisible_d():
    _require_hip()
    n, d = 2048, 123  # d % 4 != 0 exercises the scalar path
    X, y, w = _dense(n, d, seed=2)
    g, cnt = ops.grad_dense(X, y, w, seed=9, round_k=1, row_start=0,
                            rate=0.5)
    mask = torch.from_numpy(bernoulli_mask(9, 1, 0, n, 0.5)).cuda()
    g_ref, cnt_ref = torch_ref.grad_dense(X, y, w, mask, "lsq")
    assert cnt == cnt_ref
    assert float((g - g_ref).norm() / g_ref.norm()) < 1e-3


def test_grad_dense_rate_one_and_zero():
    _require_hip()
    n, d = 1024, 64
    X, y, w = _dense(n, d, seed=3)
    g, cnt = ops.grad_dense(X, y, w, seed=1, round_k=1, row_start=0, rate=1.0)
    assert cnt == n
    g0, cnt0 = ops.grad_dense(X, y, w, seed=1, round_k=1, row_start=0,
                              rate=0.0)
    assert cnt0 == 0
    assert float(g0.abs().max()) == 0.0


def test_kernel_philox_matches_numpy():
    """The in-kernel mask must be bit-identical to the numpy mask: compare
    sampled counts AND the gradient (any row disagreement shifts g)."""
    _require_hip()
    n, d = 100_000, 32
    X, y, w = _dense(n, d, seed=4)
    for rate in (0.01, 0.1, 0.5):
        g, cnt = ops.grad_dense(X, y, w, seed=123456789, round_k=77,
                                row_start=40_000, rate=rate)
        m = bernoulli_mask(123456789, 77, 40_000, n, rate)
        g_ref, cnt_ref = torch_ref.grad_dense(
            X, y, w, torch.from_numpy(m).cuda(), "lsq")
        assert cnt == cnt_ref == int(m.sum())
        assert float((g - g_ref).norm() / (g_ref.norm() + 1e-12)) < 1e-3


@pytest.mark.parametrize("objective", ["lsq", "logistic"])
def test_grad_csr_matches_ref(objective):
    _require_hip()
    n, d = 3000, 500
    indptr, indices, values, y = synthetic_csr(n, d, nnz_per_row=40, seed=5,
                                               device="cuda")
    w = torch.randn(d, device="cuda",
                    generator=torch.Generator(device="cuda").manual_seed(6))
    g, cnt = ops.grad_csr(indptr, indices, values, y, w, seed=11, round_k=3,
                          row_start=500, rate=0.4, objective=objective)
    mask = torch.from_numpy(bernoulli_mask(11, 3, 500, n, 0.4)).cuda()
    g_ref, cnt_ref = torch_ref.grad_csr(indptr, indices, values, y, w, mask,
                                        objective)
    assert cnt == cnt_ref
    assert float((g - g_ref).norm() / (g_ref.norm() + 1e-12)) < 1e-3


def test_saga_dense_matches_ref():
    _require_hip()
    n, d = 4096, 256
    X, y, w = _dense(n, d, seed=7)
    alpha = torch.randn(n, device="cuda",
                        generator=torch.Generator(device="cuda").manual_seed(8))
    g, idx, e, cnt = ops.saga_grad_dense(X, y, w, alpha, seed=21, round_k=5,
                                         row_start=0, rate=0.2)
    mask = torch.from_numpy(bernoulli_mask(21, 5, 0, n, 0.2)).cuda()
    g_ref, idx_ref, e_ref, cnt_ref = torch_ref.saga_grad_dense(
        X, y, w, alpha, mask, "lsq")
    assert cnt == cnt_ref
    # kernel emits (idx, e) in nondeterministic order — compare as sets
    order = torch.argsort(idx)
    assert torch.equal(idx[order], idx_ref)
    assert torch.allclose(e[order], e_ref, atol=1e-3, rtol=1e-3)
    assert float((g - g_ref).norm() / (g_ref.norm() + 1e-12)) < 1e-3
    # commit then recompute: corrected gradient must change accordingly
    ops.saga_commit(alpha, idx, e)
    assert torch.allclose(alpha[idx], e, atol=1e-6)


def test_saga_csr_matches_ref():
    _require_hip()
    n, d = 2000, 300
    indptr, indices, values, y = synthetic_csr(n, d, nnz_per_row=20, seed=9,
                                               device="cuda")
    w = torch.randn(d, device="cuda",
                    generator=torch.Generator(device="cuda").manual_seed(10))
    alpha = torch.zeros(n, device="cuda")
    g, idx, e, cnt = ops.saga_grad_csr(indptr, indices, values, y, w, alpha,
                                       seed=31, round_k=2, row_start=100,
                                       rate=0.3)
    mask = torch.from_numpy(bernoulli_mask(31, 2, 100, n, 0.3)).cuda()
    g_ref, idx_ref, e_ref, cnt_ref = torch_ref.saga_grad_csr(
        indptr, indices, values, y, w, alpha, mask, "lsq")
    assert cnt == cnt_ref
    order = torch.argsort(idx)
    assert torch.equal(idx[order], idx_ref)
    assert torch.allclose(e[order], e_ref, atol=1e-3, rtol=1e-3)
    assert float((g - g_ref).norm() / (g_ref.norm() + 1e-12)) < 1e-3


def test_update_kernels_match_ref():
    _require_hip()
    d = 4096
    gen = torch.Generator(device="cuda").manual_seed(12)
    w = torch.randn(d, device="cuda", generator=gen)
    g = torch.randn(d, device="cuda", generator=gen)
    ab = torch.randn(d, device="cuda", generator=gen)
    w_ref, ab_ref = w.clone(), ab.clone()
    ops.sgd_update(w, g, gamma_k=0.123, inv_batch=0.01)
    torch_ref.sgd_update(w_ref, g, 0.123, 0.01)
    assert torch.allclose(w, w_ref, atol=1e-6)
    ops.saga_update(w, g, ab, gamma=0.2, inv_batch=0.05, inv_N=0.001)
    torch_ref.saga_update(w_ref, g, ab_ref, 0.2, 0.05, 0.001)
    assert torch.allclose(w, w_ref, atol=1e-6)
    assert torch.allclose(ab, ab_ref, atol=1e-6)


def test_native_extension_is_loaded():
    """Guard against silent eager fallback: on a GPU box the in-tree .so
    must actually be importable and used."""
    _require_hip()
    import asyncframework_amd._hip_core as core
    assert getattr(core, "__hip__", False)
    assert "asyncframework_amd" in core.__file__


@pytest.mark.parametrize("objective", ["lsq", "logistic"])
def test_objective_sweep_gpu_matches_cpu_fp64(objective):
    """K7 (the GEMM-shaped objective sweep) goes through the library GEMM
    (hipBLASLt via torch.matmul) — its GPU result must match the CPU fp64
    reference within GEMM tolerance."""
    _require_hip()
    X, y, _ = _dense(4096, 256, seed=5)
    W = torch.randn(3, 256, device="cuda") * 0.1
    got = ops.objective_sweep(X, y, W, objective).cpu()
    want = torch_ref.objective_sweep(X.cpu(), y.cpu(), W.cpu(), objective)
    assert torch.allclose(got, want, rtol=1e-4, atol=1e-6), (got, want)
    # bf16 X: looser tolerance, same ballpark
    got16 = ops.objective_sweep(X.to(torch.bfloat16), y, W, objective).cpu()
    assert torch.allclose(got16, want, rtol=5e-2), (got16, want)


def test_spill_refresh_gathers_exactly_the_mask():
    """Host-spill staging refresh: the device staging table must receive the
    pinned master's values at EXACTLY the round's Philox-sampled rows (the
    same mask the gradient kernel computes) and nothing else. Also proves
    ROCm pinned host memory is device-readable (the design assumption)."""
    _require_hip()
    n, seed, rk, rs, rate = 50_000, 42, 7, 0, 0.02
    master = (torch.arange(n, dtype=torch.float32) + 1.0).pin_memory()
    a_dev = torch.zeros(n, dtype=torch.float32, device="cuda")
    y = torch.zeros(n, dtype=torch.float32, device="cuda")
    cap = int(rate * n * 2) + 4096
    rows = torch.empty(cap, dtype=torch.int32, device="cuda")
    ylist = torch.empty(cap, dtype=torch.float32, device="cuda")
    cnt = torch.zeros(1, dtype=torch.int32, device="cuda")
    ops.spill_refresh(a_dev, master, y, rows, ylist, cnt, cap,
                      seed=seed, round_k=rk, row_start=rs, rate=rate)
    torch.cuda.synchronize()
    mask = torch.from_numpy(bernoulli_mask(seed, rk, rs, n, rate))
    got = a_dev.cpu()
    assert int(cnt.item()) == int(mask.sum())
    assert torch.equal(got[mask], master[mask])
    assert torch.all(got[~mask] == 0)


def test_saga_commit_pinned_scatters_to_host():
    """Device-kernel scatter into the pinned-host master table (replaces the
    round-1 two-sync-D2H commit)."""
    _require_hip()
    n = 10_000
    master = torch.zeros(n, dtype=torch.float32).pin_memory()
    idx = torch.tensor([3, 9999, 512, 7], dtype=torch.int32, device="cuda")
    e = torch.tensor([1.5, -2.0, 3.25, 0.5], device="cuda")
    ops.saga_commit_pinned(master, idx, e)
    torch.cuda.synchronize()
    want = torch.zeros(n)
    want[idx.cpu().long()] = e.cpu()
    assert torch.equal(master, want)
