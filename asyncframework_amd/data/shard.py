"""Row-range sharding — the one-time replacement for the reference's
``repartition(numPart)`` shuffle at load (reference SparkASGDThread.scala:76,
SortShuffleManager path). Each worker owns a contiguous row range;
no runtime shuffle machinery exists or is needed."""

from __future__ import annotations

from typing import List, Tuple


def row_shards(n_rows: int, n_shards: int) -> List[Tuple[int, int]]:
    """Split [0, n_rows) into n_shards contiguous (start, stop) ranges.
    Boundaries are aligned to multiples of 4 (the Philox mask decides 4
    consecutive rows per counter — utils/philox.bernoulli_mask — and the HIP
    scan requires 4-aligned shard starts); the last shard absorbs the
    remainder."""
    base, rem = divmod(n_rows, n_shards)
    out = []
    s = 0
    for i in range(n_shards):
        t = s + base + (1 if i < rem else 0)
        if i < n_shards - 1:
            t = (t // 4) * 4
        else:
            t = n_rows
        out.append((s, t))
        s = t
    return out
