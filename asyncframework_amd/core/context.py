"""Driver-side mailbox + worker-state registry for one async optimization run.

API-compatible rebuild of the reference's three driver-side classes:

* ``ASYNCcontext``  — reference core/src/main/scala/org/apache/spark/rdd/
  ASYNCcontext.scala:14-81 (``ResultList`` LinkedBlockingQueue, ``STAT`` map,
  logical clock ``CurrentTime``/``LastTime``, blocking collect verbs).
* ``RDDPartialRes`` — reference .../rdd/RDDPartialRes.scala:13-37.
* ``workerState``   — reference .../rdd/workerState.scala:14-87.

The rebuild keeps the exact method names so a user of the reference finds the
same verbs; internally it is a plain thread-safe Python object (the Spark RDD
machinery behind it is replaced by the MI355X engine in
``asyncframework_amd.engine``).
"""

from __future__ import annotations

import queue
import threading
from typing import Dict, Generic, Optional, TypeVar

T = TypeVar("T")


class RDDPartialRes(Generic[T]):
    """Immutable envelope for one worker's reduced partial result.

    Fields mirror reference RDDPartialRes.scala:13-37:
    ``data`` (the reduced gradient), ``ts`` (staleness or submit timestamp —
    the reference packs computed staleness for ASYNCreduce, RDD.scala:1151,
    but the raw submit clock for ASYNCaggregate, RDD.scala:1333; both are
    representable here), ``recs`` (records processed; Int.MinValue when record
    counting is off, as in the reference), ``id`` (worker id).
    """

    INT_MIN = -(2 ** 31)

    __slots__ = ("data", "ts", "recs", "id")

    def __init__(self, data: T, ts: int, recs: int = INT_MIN, id: int = 0):
        self.data = data
        self.ts = ts
        self.recs = recs
        self.id = id

    def getWorkerID(self) -> int:
        return self.id

    def getbatchSize(self) -> int:
        return self.recs

    def getStaleness(self) -> int:
        return self.ts

    def gettaskResult(self) -> T:
        return self.data

    def __repr__(self) -> str:  # pragma: no cover - debug aid
        return f"RDDPartialRes(id={self.id}, ts={self.ts}, recs={self.recs})"


class workerState(Generic[T]):
    """Per-worker state: staleness, average task time (ms), availability,
    task count. Reference workerState.scala:14-87; the aggregate queries
    ``getAvailableWorkers``/``getMaxStaleness`` scan the whole STAT table,
    as in the reference (workerState.scala:63-82)."""

    def __init__(self, AC: "ASYNCcontext[T]", stale: int = 0,
                 time: int = 0, avail: bool = False):
        self._AC = AC
        self._staleness = stale
        self._averageTaskTime = time
        self._availability = avail
        self._numTasks = 0

    def updateNumTasks(self, n: int) -> None:
        self._numTasks += n

    def setStaleness(self, s: int) -> None:
        self._staleness = s

    def setAverageTaskTime(self, t: int) -> None:
        self._averageTaskTime = t

    def setAvailability(self, a: bool) -> None:
        self._availability = a

    def getNumTasks(self) -> int:
        return self._numTasks

    def getAvailability(self) -> bool:
        return self._availability

    def getStaleness(self) -> int:
        return self._staleness

    def getAverageTaskTime(self) -> int:
        return self._averageTaskTime

    def getAvailableWorkers(self) -> int:
        """Number of workers whose availability bit is set
        (reference workerState.scala:63-72)."""
        return sum(1 for s in self._AC.STAT.values() if s.getAvailability())

    def getMaxStaleness(self) -> int:
        """Max staleness over the STAT table, -1 if empty
        (reference workerState.scala:74-82)."""
        n = -1
        for s in self._AC.STAT.values():
            if s.getStaleness() > n:
                n = s.getStaleness()
        return n


class ASYNCcontext(Generic[T]):
    """Result mailbox + state registry for one asynchronous run.

    Reference ASYNCcontext.scala:14-81. ``ResultList`` is a thread-safe
    blocking queue filled by the engine's completion path (the MI355X analog
    of JobWaiter.taskSucceeded -> mergeResult, reference RDD.scala:1144-1165)
    and drained by the updater thread via ``ASYNCcollect``/``ASYNCcollectAll``.
    """

    def __init__(self) -> None:
        self.ResultList: "queue.Queue[RDDPartialRes[T]]" = queue.Queue()
        self.STAT: Dict[int, workerState[T]] = {}
        self._exactRec = False
        self._CurrentTime = 0
        self._LastTime = -(2 ** 31)
        self._lock = threading.Lock()

    # -- logical clock (reference ASYNCcontext.scala:24-54) ------------------
    def setCurrentTime(self, time: int) -> None:
        with self._lock:
            self._CurrentTime = time

    def add2currentTime(self, t: int) -> None:
        with self._lock:
            self._CurrentTime += t

    def getCurrentTime(self) -> int:
        return self._CurrentTime

    def setRecordStat(self, b: bool) -> None:
        self._exactRec = b

    def getRecordStat(self) -> bool:
        return self._exactRec

    def setLastTime(self, time: int) -> None:
        self._LastTime = time

    def isOld(self) -> bool:
        return self._CurrentTime == self._LastTime

    # -- blocking consume verbs (reference ASYNCcontext.scala:56-71) ---------
    def ASYNCcollect(self, timeout: Optional[float] = None) -> T:
        return self.ResultList.get(timeout=timeout).gettaskResult()

    def ASYNCcollectAll(self, timeout: Optional[float] = None) -> RDDPartialRes[T]:
        return self.ResultList.get(timeout=timeout)

    def getSize(self) -> int:
        return self.ResultList.qsize()

    def hasNext(self) -> bool:
        return not self.ResultList.empty()

    # -- future-returning collect (analog of Spark's AsyncRDDActions /
    #    FutureAction pattern, reference rdd/AsyncRDDActions.scala:33-137) ---
    def ASYNCcollectAsync(self) -> "futures.Future[RDDPartialRes[T]]":
        """Non-blocking collect: returns a Future fulfilled with the next
        mailbox entry."""
        from concurrent import futures
        fut: "futures.Future[RDDPartialRes[T]]" = futures.Future()

        def _wait():
            try:
                fut.set_result(self.ResultList.get())
            except Exception as e:  # pragma: no cover
                fut.set_exception(e)

        threading.Thread(target=_wait, daemon=True).start()
        return fut

    # -- engine-side producer (analog of the mergeResult trampolines,
    #    reference ASYNCcontext.scala:77-80) ---------------------------------
    def put(self, res: RDDPartialRes[T]) -> None:
        self.ResultList.put(res)
