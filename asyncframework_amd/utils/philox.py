"""Counter-based Philox4x32-10 RNG — the shared-seed sampling primitive.

The reference reproduces a Bernoulli sample identically on driver and workers
by re-running a seeded sampler with the same seed (BernoulliSampler,
reference core/.../util/random/RandomSampler.scala:144; shared-seed replay
SparkASAGAThread.scala:372-376). The MI355X rebuild replaces that with a
counter-based RNG keyed on (seed, round, global_row_index): any party —
CPU reference here, or the HIP kernel (csrc/philox.h) — derives the exact
same per-row decision with zero communication.

Counter layout: (c0,c1,c2,c3) = (row_lo, row_hi, round, 0); key = (seed_lo,
seed_hi). Row i is sampled iff out.x < rate * 2^32.
"""

from __future__ import annotations

import numpy as np

_M0 = np.uint64(0xD2511F53)
_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)


def philox4x32_10(c0, c1, c2, c3, k0, k1):
    """Vectorized Philox4x32 with 10 rounds. Inputs are uint32 numpy arrays
    (or scalars); returns the 4 output words as uint32 arrays."""
    c0 = np.asarray(c0, dtype=np.uint32)
    c1 = np.asarray(c1, dtype=np.uint32)
    c2 = np.asarray(c2, dtype=np.uint32)
    c3 = np.asarray(c3, dtype=np.uint32)
    k0 = np.uint32(k0)
    k1 = np.uint32(k1)
    with np.errstate(over="ignore"):
        for r in range(10):
            p0 = _M0 * c0.astype(np.uint64)
            p1 = _M1 * c2.astype(np.uint64)
            hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
            lo0 = p0.astype(np.uint32)
            hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
            lo1 = p1.astype(np.uint32)
            c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
            k0 = np.uint32(k0 + _W0)
            k1 = np.uint32(k1 + _W1)
    return c0, c1, c2, c3


def bernoulli_mask(seed: int, round_k: int, row_start: int, n_rows: int,
                   rate: float) -> np.ndarray:
    """Boolean sample mask for rows [row_start, row_start+n_rows) at round
    ``round_k`` — deterministic in (seed, round, absolute row index) so every
    worker/server agrees without communication (the reference's shared-seed
    trick, SparkASAGAThread.scala:372-376).

    One Philox eval decides FOUR consecutive rows (counter = absolute row
    block ``row // 4``; output word ``row % 4``) — quarters the in-kernel
    mask-scan cost, which is the dominant fixed cost per round on the
    mnist8m shape (measured ~tens of us for an 8.1M-row scan: 32-bit integer
    multiplies are slow on the VALU)."""
    if rate >= 1.0:
        return np.ones(n_rows, dtype=bool)
    b0 = row_start >> 2
    b1 = (row_start + n_rows - 1) >> 2
    blocks = np.arange(b0, b1 + 1, dtype=np.uint64)
    nb = blocks.shape[0]
    c0 = (blocks & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    c1 = (blocks >> np.uint64(32)).astype(np.uint32)
    c2 = np.full(nb, np.uint32(round_k & 0xFFFFFFFF), dtype=np.uint32)
    c3 = np.zeros(nb, dtype=np.uint32)
    k0 = np.uint32(seed & 0xFFFFFFFF)
    k1 = np.uint32((seed >> 32) & 0xFFFFFFFF)
    x0, x1, x2, x3 = philox4x32_10(c0, c1, c2, c3, k0, k1)
    allx = np.stack([x0, x1, x2, x3], axis=1).reshape(-1)
    off = row_start - 4 * b0
    vals = allx[off:off + n_rows]
    threshold = np.uint32(min(int(rate * 2 ** 32), 2 ** 32 - 1))
    return vals < threshold


def uniform01(seed: int, round_k: int, stream: int, n: int) -> np.ndarray:
    """n uniforms in [0,1) from counters (i, 0, round, stream) — used by the
    cloud-long-tail delay injector so the straggler draw is reproducible."""
    idx = np.arange(n, dtype=np.uint32)
    c2 = np.full(n, np.uint32(round_k & 0xFFFFFFFF), dtype=np.uint32)
    c3 = np.full(n, np.uint32(stream & 0xFFFFFFFF), dtype=np.uint32)
    k0 = np.uint32(seed & 0xFFFFFFFF)
    k1 = np.uint32((seed >> 32) & 0xFFFFFFFF)
    x0, _, _, _ = philox4x32_10(idx, np.zeros(n, np.uint32), c2, c3, k0, k1)
    return x0.astype(np.float64) / 2.0 ** 32
