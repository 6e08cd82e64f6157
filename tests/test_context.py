"""ASYNCcontext / workerState / RDDPartialRes API-compat tests
(reference has ZERO tests for these — SURVEY §4; we do better)."""

import threading

from asyncframework_amd import ASYNCcontext, RDDPartialRes, workerState


def test_partial_res_getters():
    r = RDDPartialRes("grad", 3, 17, 5)
    assert r.gettaskResult() == "grad"
    assert r.getStaleness() == 3
    assert r.getbatchSize() == 17
    assert r.getWorkerID() == 5


def test_partial_res_default_recs_is_int_min():
    r = RDDPartialRes("g", 0)
    assert r.getbatchSize() == -(2 ** 31)


def test_clock_semantics():
    ac = ASYNCcontext()
    assert ac.getCurrentTime() == 0
    ac.add2currentTime(3)
    assert ac.getCurrentTime() == 3
    ac.setCurrentTime(10)
    assert ac.getCurrentTime() == 10
    assert not ac.isOld()
    ac.setLastTime(10)
    assert ac.isOld()


def test_record_stat_flag():
    ac = ASYNCcontext()
    assert not ac.getRecordStat()
    ac.setRecordStat(True)
    assert ac.getRecordStat()


def test_mailbox_blocking_collect():
    ac = ASYNCcontext()
    assert ac.getSize() == 0
    assert not ac.hasNext()
    out = []

    def producer():
        ac.put(RDDPartialRes(42, 1, 0, 2))

    t = threading.Thread(target=producer)
    t.start()
    res = ac.ASYNCcollectAll(timeout=5)
    t.join()
    assert res.gettaskResult() == 42
    assert res.getWorkerID() == 2
    assert ac.getSize() == 0


def test_asynccollect_returns_payload():
    ac = ASYNCcontext()
    ac.put(RDDPartialRes("payload", 0, 0, 0))
    assert ac.ASYNCcollect(timeout=1) == "payload"


def test_worker_state_aggregates():
    ac = ASYNCcontext()
    for i in range(4):
        ac.STAT[i] = workerState(ac)
    assert ac.STAT[0].getAvailableWorkers() == 0
    ac.STAT[1].setAvailability(True)
    ac.STAT[3].setAvailability(True)
    assert ac.STAT[0].getAvailableWorkers() == 2
    ac.STAT[2].setStaleness(7)
    assert ac.STAT[0].getMaxStaleness() == 7
    ac.STAT[2].updateNumTasks(3)
    assert ac.STAT[2].getNumTasks() == 3
