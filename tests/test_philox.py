"""Philox shared-seed sampling: determinism and rate — the reference's
driver/worker seed-agreement trick (SparkASAGAThread.scala:372-376) must be
reproducible with zero communication."""

import numpy as np

from asyncframework_amd.utils.philox import (bernoulli_mask, philox4x32_10,
                                             uniform01)


def test_philox_deterministic():
    a = philox4x32_10(np.uint32(1), np.uint32(2), np.uint32(3), np.uint32(4),
                      0xDEADBEEF, 0x12345678)
    b = philox4x32_10(np.uint32(1), np.uint32(2), np.uint32(3), np.uint32(4),
                      0xDEADBEEF, 0x12345678)
    for x, y in zip(a, b):
        assert np.all(x == y)


def test_philox_counter_sensitivity():
    base = philox4x32_10(np.uint32(0), np.uint32(0), np.uint32(0),
                         np.uint32(0), 42, 0)[0]
    for c in range(4):
        args = [np.uint32(0)] * 4
        args[c] = np.uint32(1)
        other = philox4x32_10(*args, 42, 0)[0]
        assert other != base


def test_mask_shard_consistency():
    """A worker computing its shard's mask must agree with a global mask —
    masks are keyed by ABSOLUTE row index."""
    full = bernoulli_mask(seed=7, round_k=3, row_start=0, n_rows=1000,
                          rate=0.3)
    part = bernoulli_mask(seed=7, round_k=3, row_start=400, n_rows=100,
                          rate=0.3)
    assert np.array_equal(full[400:500], part)


def test_mask_rate():
    m = bernoulli_mask(seed=1, round_k=0, row_start=0, n_rows=200_000,
                       rate=0.1)
    frac = m.mean()
    assert abs(frac - 0.1) < 0.005


def test_mask_varies_with_round():
    m1 = bernoulli_mask(seed=1, round_k=1, row_start=0, n_rows=1000, rate=0.5)
    m2 = bernoulli_mask(seed=1, round_k=2, row_start=0, n_rows=1000, rate=0.5)
    assert not np.array_equal(m1, m2)


def test_mask_rate_one_is_all():
    m = bernoulli_mask(seed=1, round_k=0, row_start=0, n_rows=100, rate=1.0)
    assert m.all()


def test_uniform01_range():
    u = uniform01(seed=9, round_k=2, stream=1, n=10_000)
    assert (u >= 0).all() and (u < 1).all()
    assert abs(u.mean() - 0.5) < 0.02


def _philox_scalar_ref(c0, c1, c2, c3, k0, k1):
    """Independent pure-Python Philox4x32-10 (no numpy vectorization) —
    double-checks the vectorized implementation and, transitively, the HIP
    kernel (which is tested bit-identical against the numpy one)."""
    M = (1 << 32) - 1
    for _ in range(10):
        p0 = (0xD2511F53 * c0) & ((1 << 64) - 1)
        p1 = (0xCD9E8D57 * c2) & ((1 << 64) - 1)
        hi0, lo0 = p0 >> 32, p0 & M
        hi1, lo1 = p1 >> 32, p1 & M
        c0, c1, c2, c3 = (hi1 ^ c1 ^ k0) & M, lo1, (hi0 ^ c3 ^ k1) & M, lo0
        k0 = (k0 + 0x9E3779B9) & M
        k1 = (k1 + 0xBB67AE85) & M
    return c0, c1, c2, c3


def test_vectorized_matches_scalar_reference():
    import random
    rng = random.Random(7)
    for _ in range(20):
        c = [rng.getrandbits(32) for _ in range(4)]
        k = [rng.getrandbits(32) for _ in range(2)]
        vec = philox4x32_10(np.uint32(c[0]), np.uint32(c[1]),
                            np.uint32(c[2]), np.uint32(c[3]), k[0], k[1])
        ref = _philox_scalar_ref(*c, *k)
        assert tuple(int(x) for x in vec) == ref


def test_mask_matches_scalar_reference():
    seed, rk, rate = 987654321, 12, 0.37
    m = bernoulli_mask(seed, rk, row_start=8, n_rows=40, rate=rate)
    thr = min(int(rate * 2 ** 32), 2 ** 32 - 1)
    k0, k1 = seed & 0xFFFFFFFF, (seed >> 32) & 0xFFFFFFFF
    for i, row in enumerate(range(8, 48)):
        blk = row >> 2
        out = _philox_scalar_ref(blk & 0xFFFFFFFF, blk >> 32, rk, 0, k0, k1)
        assert bool(m[i]) == (out[row % 4] < thr)
