"""The user-facing ASYNC verb layer (core/rdd.py): a reference driver's code
shape must run unchanged against it — this file replays the §3.2 driver loop
(SparkASGDThread.scala:230-345 + updater :153-226) in Python and checks the
reference's documented quirks."""

import threading

import numpy as np
import pytest

from asyncframework_amd.core.context import ASYNCcontext
from asyncframework_amd.core.rdd import INT_MIN, AsyncRDD, ASYNCbroadcast


@pytest.fixture(autouse=True)
def _fresh_statics():
    AsyncRDD.reset_statics()
    ASYNCbroadcast.reset_registry()
    yield
    AsyncRDD.reset_statics()
    ASYNCbroadcast.reset_registry()


def _make_points(n=200, d=8, P=4, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, d))
    w_true = rng.standard_normal(d)
    y = X @ w_true
    pts = [(X[i], y[i]) for i in range(n)]
    parts = [pts[i * (n // P):(i + 1) * (n // P)] for i in range(P)]
    return AsyncRDD(parts), X, y


def test_first_async_call_blocks_and_seeds_workerlist():
    rdd, X, y = _make_points(P=4)
    AC = ASYNCcontext()
    assert AsyncRDD.Init
    rdd.map(lambda p: 1).ASYNCreduce(lambda a, b: a + b, AC)
    # first call blocks (DAGScheduler first_iter warm-up): results present
    assert AC.getSize() == 4
    assert AsyncRDD.WorkerList == [0, 1, 2, 3]
    assert not AsyncRDD.Init and not AsyncRDD.first_iter
    counts = sorted(AC.ASYNCcollectAll().gettaskResult() for _ in range(4))
    assert sum(counts) == 200


def test_empty_workerlist_is_noop():
    rdd, _, _ = _make_points(P=2)
    AC = ASYNCcontext()
    AsyncRDD.Init = False
    AsyncRDD.WorkerList = []
    rdd.ASYNCreduce(lambda a, b: a, AC)
    assert AC.getSize() == 0 and not AC.hasNext()


def test_recs_is_int_min():
    rdd, _, _ = _make_points(P=2)
    AC = ASYNCcontext()
    rdd.map(lambda p: 0).ASYNCreduce(lambda a, b: a, AC)
    pr = AC.ASYNCcollectAll()
    assert pr.getbatchSize() == INT_MIN


def test_asyncaggregate_packs_raw_ts():
    """ASGD packs computed staleness (RDD.scala:1151); ASAGA packs the raw
    submit clock (RDD.scala:1333)."""
    rdd, _, _ = _make_points(P=2)
    AC = ASYNCcontext()
    AC.setCurrentTime(7)
    rdd.ASYNCaggregate(0, lambda acc, p: acc + 1, lambda a, b: a + b, AC)
    for _ in range(2):
        assert AC.ASYNCcollectAll().getStaleness() == 7  # raw ts
    AC2 = ASYNCcontext()
    AC2.setCurrentTime(7)
    rdd2, _, _ = _make_points(P=2)
    AsyncRDD.reset_statics()
    rdd2.map(lambda p: 0).ASYNCreduce(lambda a, b: a, AC2)
    st = sorted(AC2.ASYNCcollectAll().getStaleness() for _ in range(2))
    assert st == [0, 1]  # CurrentTime - ts, clock bumps between merges


def test_barrier_excludes_unavailable_workers():
    rdd, _, _ = _make_points(P=4)
    AC = ASYNCcontext()
    rdd.map(lambda p: 1).ASYNCreduce(lambda a, b: a + b, AC)  # seeds STAT
    while AC.hasNext():
        AC.ASYNCcollectAll()
    AC.STAT[2].setAvailability(False)
    filtered = rdd.ASYNCbarrier(lambda st: st.getAvailability(), AC.STAT)
    assert AsyncRDD.WorkerList == [0, 1, 3]
    filtered.map(lambda p: 1).ASYNCreduce(lambda a, b: a + b, AC)
    ids = set()
    while AC.hasNext():
        ids.add(AC.ASYNCcollectAll().getWorkerID())
    assert ids == {0, 1, 3}


def test_barrier_includes_partitions_without_stat_entry():
    rdd, _, _ = _make_points(P=3)
    AC = ASYNCcontext()  # empty STAT: everyone included (RDD.scala:1062)
    rdd.ASYNCbarrier(lambda st: st.getAvailability(), AC.STAT)
    assert AsyncRDD.WorkerList == [0, 1, 2]


def test_sample_matches_engine_philox():
    """The verb layer's sample() and the kernels' mask agree — the shared
    seed trick spans both layers."""
    from asyncframework_amd.utils.philox import bernoulli_mask
    rdd, _, _ = _make_points(n=200, P=4)
    k, seed = 3, 42
    tagged = rdd.zipWithIndex().sample(False, 0.5, seed + k + 1)
    kept = [i for (_, i) in tagged.collect()]
    m = bernoulli_mask(seed=seed + k + 1, round_k=0, row_start=0,
                       n_rows=200, rate=0.5)
    assert kept == [i for i in range(200) if m[i]]


def test_asyncbroadcast_versioned_reads():
    b0 = ASYNCbroadcast([1.0, 2.0])
    b1 = ASYNCbroadcast([3.0, 4.0])
    assert b1.value() == [3.0, 4.0]
    assert b1.value(b0.bid) == [1.0, 2.0]  # stale read by index


def test_full_async_driver_loop_converges():
    """The §3.2 shape: gate -> broadcast -> barrier -> sample -> map(gradfun)
    -> ASYNCreduce; updater thread drains the mailbox with the tau filter and
    steps w. Objective must decrease."""
    P, d, tau, gamma, rate = 4, 8, 1 << 30, 0.05, 0.5
    rdd, X, y = _make_points(n=200, d=d, P=P, seed=1)
    AC = ASYNCcontext()
    w = np.zeros(d)
    k = 0
    num_iter = 60
    stop = threading.Event()

    def gradfun(point, wv):
        x, yy = point
        return (x @ wv - yy) * x

    def updater():
        nonlocal w, k
        while not stop.is_set():
            try:
                pr = AC.ASYNCcollectAll(timeout=0.2)
            except Exception:
                continue
            if pr.gettaskResult() is None:
                continue
            if pr.getStaleness() <= tau:
                g = pr.gettaskResult()
                w = w - gamma * g / (rate * 200 / P)
                k += 1

    th = threading.Thread(target=updater, daemon=True)
    th.start()
    import time as _t
    rounds = 0
    while k < num_iter and rounds < 500:
        # quorum gate (SparkASGDThread.scala:233-237)
        if AC.STAT:
            avail = next(iter(AC.STAT.values())).getAvailableWorkers()
            if avail < int(P * 0.5):
                _t.sleep(0.001)
                continue
        bc = ASYNCbroadcast(w.copy())
        filtered = rdd.ASYNCbarrier(
            lambda st: st.getAvailability(), AC.STAT)
        sampled = filtered.sample(False, rate, 42 + rounds + 1)
        grads = sampled.map(lambda p, _bc=bc: gradfun(p, _bc.value()))
        grads.ASYNCreduce(lambda a, b: a + b, AC)
        rounds += 1
    stop.set()
    th.join(timeout=5)
    obj0 = float(((X @ np.zeros(d) - y) ** 2).mean())
    obj1 = float(((X @ w - y) ** 2).mean())
    assert k >= num_iter
    assert obj1 < obj0 * 0.5


def test_mappartitions_delay_injection_shape():
    """The reference injects stragglers via a mapPartitions sleep
    (SparkASGDThread.scala:287-312) — same code shape works here."""
    import time as _t
    rdd, _, _ = _make_points(n=40, P=4)
    AC = ASYNCcontext()
    slept = []

    def inject(elems):
        slept.append(1)
        return elems

    delayed = rdd.mapPartitions(inject)
    delayed.map(lambda p: 1).ASYNCreduce(lambda a, b: a + b, AC)
    total = sum(AC.ASYNCcollectAll().gettaskResult() for _ in range(4))
    assert total == 40
    assert len(slept) == 4  # ran once per partition


def test_mappartitions_with_index():
    rdd, _, _ = _make_points(n=40, P=4)
    out = rdd.mapPartitionsWithIndex(
        lambda pid, elems: [pid] if pid % 2 == 0 else []).collect()
    assert out == [0, 2]


def test_async_actions_futures():
    """AsyncRDDActions analog (reference AsyncRDDActions.scala:33-137):
    future-returning count/collect/foreach."""
    rdd, _, _ = _make_points(n=40, P=4)
    assert rdd.countAsync().result(timeout=30) == 40
    assert len(rdd.collectAsync().result(timeout=30)) == 40
    seen = []
    rdd.map(lambda p: 1).foreachAsync(seen.append).result(timeout=30)
    assert len(seen) == 40
    # composes with transforms
    assert rdd.sample(False, 0.5, 9).countAsync().result(timeout=30) <= 40
