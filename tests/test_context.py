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


def test_server_stat_running_average():
    """on_completion maintains the per-worker running average task time
    (reference workerState avg + mergeResult refresh, RDD.scala:1148-1153)."""
    import torch

    from asyncframework_amd.engine.config import EngineConfig
    from asyncframework_amd.engine.messages import WorkerResult
    from asyncframework_amd.engine.server import Server
    cfg = EngineConfig(d=4, N=8, num_workers=2)
    srv = Server(cfg, device=torch.device("cpu"))
    g = torch.zeros(4)
    srv.on_completion(WorkerResult(worker_id=0, g=g, ts=0, k_submit=0,
                                   nrows=1, elapsed_ms=100.0))
    assert srv.AC.STAT[0].getAverageTaskTime() == 100
    srv.on_completion(WorkerResult(worker_id=0, g=g, ts=1, k_submit=1,
                                   nrows=1, elapsed_ms=200.0))
    assert srv.AC.STAT[0].getAverageTaskTime() == 150
    assert srv.AC.STAT[0].getNumTasks() == 2
    assert srv.AC.getCurrentTime() == 2
    # second worker untouched
    assert srv.AC.STAT[1].getNumTasks() == 0
