"""Property-based tests (hypothesis) for the invariants everything else
rests on — the analog of stock Spark's ScalaCheck property suites
(the substrate's test pyramid is the model; the shared-seed
sampling determinism is called out as 'subtle and must be tested')."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from asyncframework_amd.data.shard import row_shards
from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.engine.delay import DelayInjector
from asyncframework_amd.engine.messages import (HDR, Dispatch, WorkerResult,
                                                pack_dispatch, pack_result,
                                                unpack_dispatch,
                                                unpack_result)
from asyncframework_amd.utils.philox import bernoulli_mask


@settings(max_examples=60, deadline=None)
@given(seed=st.integers(0, 2 ** 63 - 1),
       round_k=st.integers(0, 2 ** 31 - 1),
       start4=st.integers(0, 5000),
       n_rows=st.integers(1, 2000),
       rate=st.floats(0.0, 1.0))
def test_mask_shard_consistency_property(seed, round_k, start4, n_rows, rate):
    """Any 4-aligned window of the global mask equals the window-local mask:
    the driver/worker zero-communication agreement (the reference's shared
    seed trick) holds for every (seed, round, shard, rate)."""
    row_start = start4 * 4
    part = bernoulli_mask(seed=seed, round_k=round_k, row_start=row_start,
                          n_rows=n_rows, rate=rate)
    full = bernoulli_mask(seed=seed, round_k=round_k, row_start=0,
                          n_rows=row_start + n_rows, rate=rate)
    assert np.array_equal(full[row_start:row_start + n_rows], part)


@settings(max_examples=100, deadline=None)
@given(n=st.integers(1, 100_000), p=st.integers(1, 64))
def test_row_shards_partition_property(n, p):
    """Shards cover [0, n) exactly once, in order, with 4-aligned interior
    boundaries (the Philox 4-rows-per-eval contract)."""
    shards = row_shards(n, p)
    assert len(shards) == p
    prev_end = 0
    for s, t in shards:
        assert s == prev_end
        assert s % 4 == 0
        assert t >= s
        prev_end = t
    assert prev_end == n


@settings(max_examples=60, deadline=None)
@given(d=st.integers(1, 64),
       ts=st.integers(0, 2 ** 24 - 1),
       k=st.integers(0, 2 ** 24 - 1),
       accept=st.booleans(),
       stop=st.booleans(),
       snap=st.integers(0, 2),
       delay=st.floats(0.0, 100.0, allow_nan=False))
def test_dispatch_wire_roundtrip_property(d, ts, k, accept, stop, snap,
                                          delay):
    buf = torch.zeros(d + HDR, dtype=torch.float32)
    w = torch.randn(d)
    msg = Dispatch(w=w, ts=ts, k_submit=k, accept_prev=accept, stop=stop,
                   delay_s=delay, snap=snap)
    pack_dispatch(buf, d, msg)
    out = unpack_dispatch(buf, d)
    assert (out.ts, out.k_submit, out.accept_prev, out.stop, out.snap) == \
        (ts, k, accept, stop, snap)
    assert abs(out.delay_s - delay) < 1e-3
    assert torch.equal(out.w, w)


@settings(max_examples=60, deadline=None)
@given(d=st.integers(1, 64),
       ts=st.integers(0, 2 ** 24),
       k=st.integers(0, 2 ** 24),
       nrows=st.integers(0, 2 ** 24),
       wid=st.integers(0, 1000),
       ms=st.floats(0.0, 1e6, allow_nan=False))
def test_result_wire_roundtrip_property(d, ts, k, nrows, wid, ms):
    buf = torch.zeros(d + HDR, dtype=torch.float32)
    g = torch.randn(d)
    pack_result(buf, d, WorkerResult(worker_id=wid, g=g, ts=ts, k_submit=k,
                                     nrows=nrows, elapsed_ms=ms))
    out = unpack_result(buf, d, wid)
    assert (out.worker_id, out.ts, out.k_submit, out.nrows) == \
        (wid, ts, k, nrows)
    assert torch.equal(out.g, g)


@settings(max_examples=100, deadline=None)
@given(p=st.integers(1, 256), ratio=st.floats(0.0, 1.0))
def test_gate_bounds_property(p, ratio):
    gate = EngineConfig(num_workers=p, bucket_ratio=ratio).gate
    assert 0 <= gate <= p
    assert gate <= p * ratio < gate + 1  # floor


@settings(max_examples=30, deadline=None)
@given(p=st.integers(1, 64), coeff=st.sampled_from([-1.0, 0.5, 1.0, 2.0]),
       seed=st.integers(0, 2 ** 31 - 1), k=st.integers(0, 1000))
def test_delay_nonnegative_and_deterministic_property(p, coeff, seed, k):
    inj = DelayInjector(p, coeff=coeff, seed=seed, calib_window=0)
    inj._cul_time, inj._cul_count = 80.0, 1
    inj.maybe_activate(1)
    for wid in range(min(p, 8)):
        d1 = inj.delay_ms(wid, k)
        assert d1 >= 0.0
        assert d1 == inj.delay_ms(wid, k)


@settings(max_examples=40, deadline=None)
@given(n=st.integers(1, 300), p=st.integers(1, 8),
       frac=st.floats(0.0, 1.0), seed=st.integers(0, 2 ** 31 - 1))
def test_asyncrdd_lineage_properties(n, p, frac, seed):
    """Verb-layer lineage laws: map composes elementwise, sample is a
    deterministic order-preserving subset, count is map-invariant."""
    from asyncframework_amd.core.rdd import AsyncRDD
    p = min(p, n)
    data = list(range(n))
    parts = [data[i * n // p:(i + 1) * n // p] for i in range(p)]
    rdd = AsyncRDD(parts)
    assert rdd.collect() == data
    assert rdd.map(lambda x: x * 3).collect() == [x * 3 for x in data]
    s1 = rdd.sample(False, frac, seed).collect()
    s2 = rdd.sample(False, frac, seed).collect()
    assert s1 == s2                      # deterministic
    assert s1 == [x for x in data if x in set(s1)]  # order-preserving subset
    assert rdd.map(lambda x: None).count() == n
    assert rdd.filter(lambda x: x % 2 == 0).count() == len(
        [x for x in data if x % 2 == 0])


@settings(max_examples=40, deadline=None)
@given(seed=st.integers(0, 2 ** 63 - 1), round_k=st.integers(0, 2 ** 20),
       start=st.integers(0, 9999), n_rows=st.integers(1, 500),
       rate=st.floats(0.0, 1.0))
def test_mask_window_consistency_unaligned(seed, round_k, start, n_rows,
                                           rate):
    """The numpy mask is window-consistent for ANY start (the verb layer
    uses arbitrary partition offsets; only the HIP kernel requires
    4-aligned shard starts)."""
    part = bernoulli_mask(seed=seed, round_k=round_k, row_start=start,
                          n_rows=n_rows, rate=rate)
    full = bernoulli_mask(seed=seed, round_k=round_k, row_start=0,
                          n_rows=start + n_rows, rate=rate)
    assert np.array_equal(full[start:start + n_rows], part)
