"""User-facing ASYNC verb layer — the reference's RDD extension API.

The engines (``asyncframework_amd.engine``) ARE the framework's hot path;
this module is the *compatibility surface* for driver code written against
the reference's verbs (SURVEY §2.1 "API surface to keep compatible"):

* ``AsyncRDD.ASYNCbarrier(pred, table)``  — reference RDD.scala:1050-1077
* ``AsyncRDD.ASYNCreduce(f, AC)``         — reference RDD.scala:1087-1171
* ``AsyncRDD.ASYNCaggregate(zero, seqOp, combOp, AC)`` — RDD.scala:1268-1345
* ``set_mode`` / first-call-blocks        — SparkContext.scala:91-93,
                                            DAGScheduler.scala:631-670
* ``ASYNCbroadcast``                      — broadcast/ASYNCbroadcast.scala

A partitioned collection of plain Python objects, evaluated per partition on
a shared thread pool (the Executor/TaskRunner analog); results are merged
driver-side under a lock (the synchronized JobWaiter.taskSucceeded path,
reference JobWaiter.scala:56-60) into an ``ASYNCcontext`` mailbox. Faithful
quirks reproduced on purpose:

* ``WorkerList``/``Init`` are CLASS-level statics (reference keeps them as
  statics on the RDD object, RDD.scala:2152-2153) — one async run per
  process, exactly like the reference.
* the FIRST async submit blocks once (warm-up; DAGScheduler.scala:641,657),
  later submits return immediately in mode 1.
* an empty ``WorkerList`` makes ``ASYNCreduce`` a no-op (RDD.scala:1095-1097).
* ``ASYNCreduce`` packs computed staleness ``CurrentTime - ts``
  (RDD.scala:1151) while ``ASYNCaggregate`` packs the RAW submit clock
  (RDD.scala:1333) — consumers differ accordingly (τ test forms).
* ``recs`` is always Int.MinValue (record counting is commented out in the
  reference, RDD.scala:1104-1122).

Sampling determinism: ``sample(False, b, seed)`` uses the same Philox
Bernoulli mask as the HIP kernels, keyed by the element's absolute row index
(partition offset + position) — the shared-seed replay trick
(SparkASAGAThread.scala:372-376) therefore works between this layer and the
engines/kernels.
"""

from __future__ import annotations

import copy
import threading
import time
from concurrent.futures import ThreadPoolExecutor, wait
from typing import Any, Callable, Dict, List, Optional, Sequence

from ..utils.philox import bernoulli_mask
from .context import ASYNCcontext, RDDPartialRes, workerState

INT_MIN = RDDPartialRes.INT_MIN


class AsyncRDD:
    """Partitioned collection with the reference's ASYNC verbs."""

    # reference statics (RDD.scala:2152-2153) + non-blocking mode state
    # (SparkContext.mode / DAGScheduler.first_iter)
    WorkerList: List[int] = []
    Init: bool = True
    mode: int = 0
    first_iter: bool = True

    def __init__(self, partitions: Sequence[Sequence[Any]],
                 _stages: Optional[List] = None,
                 _root: Optional["AsyncRDD"] = None):
        self._parts = [list(p) for p in partitions] if _root is None else None
        self._stages = _stages or []
        self._root = _root or self
        if _root is None:
            self._offsets = []
            off = 0
            for p in self._parts:
                self._offsets.append(off)
                off += len(p)
            self._pool = ThreadPoolExecutor(
                max_workers=max(1, len(self._parts)),
                thread_name_prefix="asyncrdd")
            self._merge_lock = threading.Lock()

    # -- reference reset hook (tests / multiple runs per process) ------------
    @classmethod
    def reset_statics(cls) -> None:
        cls.WorkerList = []
        cls.Init = True
        cls.mode = 0
        cls.first_iter = True

    @classmethod
    def set_mode(cls, m: int) -> None:
        """reference SparkContext.set_mode (SparkContext.scala:91-93)."""
        cls.mode = m

    # -- lazy transforms -----------------------------------------------------
    def _derive(self, stage) -> "AsyncRDD":
        return AsyncRDD((), _stages=self._stages + [stage], _root=self._root)

    def map(self, f: Callable[[Any], Any]) -> "AsyncRDD":
        return self._derive(("map", f))

    def filter(self, f: Callable[[Any], bool]) -> "AsyncRDD":
        return self._derive(("filter", f))

    def mapPartitions(self, f) -> "AsyncRDD":
        """Per-partition transform (iterator -> iterator) — the hook the
        reference's delay injection uses (mapPartitions sleep,
        SparkASGDThread.scala:287-312)."""
        return self._derive(("mapparts", lambda pid, elems: f(elems)))

    def mapPartitionsWithIndex(self, f) -> "AsyncRDD":
        """(partition index, iterator) -> iterator — the primitive
        ASYNCbarrier is built on in the reference (RDD.scala:1066-1073)."""
        return self._derive(("mapparts", f))

    def zipWithIndex(self) -> "AsyncRDD":
        """(element, absolute index) pairs (reference RDD.scala:1527-1528)."""
        return self._derive(("zipidx", None))

    def sample(self, withReplacement: bool, fraction: float,
               seed: int) -> "AsyncRDD":
        """Bernoulli row sampling (reference RDD.scala:488-500 →
        BernoulliSampler). Philox mask keyed by absolute row index — callers
        pass ``seed + k + 1`` per round exactly like the reference
        (SparkASGDThread.scala:314)."""
        assert not withReplacement, "only Bernoulli (withReplacement=False)"
        return self._derive(("sample", (fraction, seed)))

    def getNumPartitions(self) -> int:
        return len(self._root._parts)

    # -- per-partition evaluation (the Executor.TaskRunner analog) -----------
    def _eval_partition(self, pid: int) -> List[Any]:
        root = self._root
        elems = root._parts[pid]
        offset = root._offsets[pid]
        for kind, arg in self._stages:
            if kind == "map":
                elems = [arg(e) for e in elems]
            elif kind == "filter":
                elems = [e for e in elems if arg(e)]
            elif kind == "mapparts":
                elems = list(arg(pid, iter(elems)))
            elif kind == "zipidx":
                elems = [(e, offset + i) for i, e in enumerate(elems)]
            elif kind == "sample":
                frac, seed = arg
                m = bernoulli_mask(seed=seed, round_k=0, row_start=offset,
                                   n_rows=len(elems), rate=frac)
                elems = [e for e, keep in zip(elems, m) if keep]
            elif kind == "barrier":
                if pid not in AsyncRDD.WorkerList:
                    return []  # excluded partitions emit Iterator.empty
        return elems

    def collect(self) -> List[Any]:
        out: List[Any] = []
        for pid in range(self.getNumPartitions()):
            out.extend(self._eval_partition(pid))
        return out

    # -- ASYNC verbs ---------------------------------------------------------
    def ASYNCbarrier(self, pred: Callable[[workerState], bool],
                     table: Dict[int, workerState]) -> "AsyncRDD":
        """Rebuild the static WorkerList from the STAT table and return an
        RDD that emits nothing for excluded partitions (reference
        RDD.scala:1050-1077; partitions with no STAT entry are always
        included, :1062)."""
        AsyncRDD.WorkerList = [
            pid for pid in range(self.getNumPartitions())
            if pid not in table or pred(table[pid])]
        return self._derive(("barrier", None))

    def _submit(self, AC: ASYNCcontext, task: Callable[[int], Any],
                pack_raw_ts: bool) -> None:
        """Shared submit/merge mechanics of ASYNCreduce/ASYNCaggregate
        (RDD.scala:1100-1168, merge :1144-1165)."""
        if AsyncRDD.Init:  # first call computes on ALL partitions
            AsyncRDD.WorkerList = list(range(self.getNumPartitions()))
            AsyncRDD.Init = False
        if not AsyncRDD.WorkerList:
            return  # RDD.scala:1095-1097
        ts = AC.getCurrentTime()
        root = self._root
        for pid in AsyncRDD.WorkerList:  # mark busy (:1136-1142)
            st = AC.STAT.setdefault(pid, workerState(AC))
            st.setAvailability(False)
        AsyncRDD.set_mode(1)  # :1167

        def run_one(pid: int) -> None:
            t0 = time.perf_counter()
            result = task(pid)
            ms = int((time.perf_counter() - t0) * 1000)
            with root._merge_lock:  # JobWaiter.taskSucceeded synchronized
                packed = ts if pack_raw_ts else AC.getCurrentTime() - ts
                AC.put(RDDPartialRes(result, packed, INT_MIN, pid))
                st = AC.STAT.setdefault(pid, workerState(AC))
                st.setAvailability(True)
                st.updateNumTasks(1)
                n = st.getNumTasks()
                st.setAverageTaskTime(
                    (st.getAverageTaskTime() * (n - 1) + ms) // max(n, 1))
                st.setStaleness(AC.getCurrentTime() - ts)
                AC.add2currentTime(1)  # :1158

        futs = [root._pool.submit(run_one, pid)
                for pid in AsyncRDD.WorkerList]
        if AsyncRDD.mode == 0 or AsyncRDD.first_iter:
            wait(futs)  # DAGScheduler.scala:641,657: first async job blocks
        AsyncRDD.first_iter = False

    def ASYNCreduce(self, f: Callable[[Any, Any], Any],
                    AC: ASYNCcontext) -> None:
        """Non-blocking reduce: each listed partition folds ``f`` locally;
        the per-partition result lands in ``AC.ResultList`` with COMPUTED
        staleness (reference RDD.scala:1087-1171)."""
        def task(pid: int):
            elems = self._eval_partition(pid)
            if not elems:
                return None
            acc = elems[0]
            for e in elems[1:]:
                acc = f(acc, e)
            return acc
        self._submit(AC, task, pack_raw_ts=False)

    def ASYNCaggregate(self, zeroValue: Any, seqOp: Callable[[Any, Any], Any],
                       combOp: Callable[[Any, Any], Any],
                       AC: ASYNCcontext) -> None:
        """Non-blocking aggregate: per-partition seqOp fold from a deep-copied
        zero; packs the RAW submit clock (consumer checks ``k - ts <= tau``).
        ``combOp`` merges sub-results within a partition task (single chunk
        here, so it mirrors the reference where cross-partition combination
        happens in the updater thread). Reference RDD.scala:1268-1345."""
        def task(pid: int):
            acc = copy.deepcopy(zeroValue)
            for e in self._eval_partition(pid):
                acc = seqOp(acc, e)
            return acc
        self._submit(AC, task, pack_raw_ts=True)

    # -- blocking verbs for the shutdown epilogue (mode 0) -------------------
    def reduce(self, f: Callable[[Any, Any], Any]) -> Any:
        elems = self.collect()
        acc = elems[0]
        for e in elems[1:]:
            acc = f(acc, e)
        return acc

    def count(self) -> int:
        return sum(len(self._eval_partition(pid))
                   for pid in range(self.getNumPartitions()))

    # -- future-returning actions (the AsyncRDDActions analog, reference
    #    rdd/AsyncRDDActions.scala:33-137 — Spark's OTHER, pre-existing
    #    async mechanism; the ASYNC framework itself used mode-1 runJob) --
    def countAsync(self):
        return self._root._pool.submit(self.count)

    def collectAsync(self):
        return self._root._pool.submit(self.collect)

    def foreachAsync(self, f: Callable[[Any], None]):
        def _run():
            for e in self.collect():
                f(e)
        return self._root._pool.submit(_run)


class ASYNCbroadcast:
    """Versioned value store: ``value(index)`` can read an OLDER broadcast's
    value — the stale-weight-fetch mechanism (reference
    broadcast/ASYNCbroadcast.scala:21-27, block-id swap trick). Here the
    version ring is an in-process registry (per-dispatch weight snapshots in
    the engines play this role on GPU)."""

    _registry: List[Any] = []
    _lock = threading.Lock()

    def __init__(self, value: Any):
        with ASYNCbroadcast._lock:
            self.bid = len(ASYNCbroadcast._registry)
            ASYNCbroadcast._registry.append(value)

    def value(self, index: Optional[int] = None) -> Any:
        return ASYNCbroadcast._registry[self.bid if index is None else index]

    @classmethod
    def reset_registry(cls) -> None:
        cls._registry = []
