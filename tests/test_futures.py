"""Future-returning collect — the AsyncRDDActions/FutureAction analog
(reference rdd/AsyncRDDActions.scala:33-137)."""

import threading
import time

from asyncframework_amd import ASYNCcontext, RDDPartialRes


def test_collect_async_resolves():
    ac = ASYNCcontext()
    fut = ac.ASYNCcollectAsync()
    assert not fut.done()
    ac.put(RDDPartialRes("grad", 1, 0, 7))
    res = fut.result(timeout=5)
    assert res.gettaskResult() == "grad"
    assert res.getWorkerID() == 7


def test_collect_async_ordering():
    ac = ASYNCcontext()
    f1 = ac.ASYNCcollectAsync()
    ac.put(RDDPartialRes("a", 0, 0, 0))
    assert f1.result(timeout=5).gettaskResult() == "a"
    ac.put(RDDPartialRes("b", 0, 0, 1))
    f2 = ac.ASYNCcollectAsync()
    assert f2.result(timeout=5).gettaskResult() == "b"
