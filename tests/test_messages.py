"""Wire-format pack/unpack roundtrip (dist engine message layer)."""

import torch

from asyncframework_amd.engine.messages import (HDR, Dispatch, WorkerResult,
                                                pack_dispatch, pack_result,
                                                unpack_dispatch,
                                                unpack_result)


def test_dispatch_roundtrip():
    d = 8
    buf = torch.zeros(d + HDR)
    w = torch.arange(d, dtype=torch.float32)
    msg = Dispatch(w=w, ts=13, k_submit=101, accept_prev=False,
                   delay_s=0.25, stop=False)
    pack_dispatch(buf, d, msg)
    out = unpack_dispatch(buf, d)
    assert torch.equal(out.w, w)
    assert out.ts == 13 and out.k_submit == 101
    assert out.accept_prev is False and out.stop is False
    assert abs(out.delay_s - 0.25) < 1e-6


def test_stop_dispatch_roundtrip():
    d = 4
    buf = torch.zeros(d + HDR)
    pack_dispatch(buf, d, Dispatch(w=None, stop=True))
    assert unpack_dispatch(buf, d).stop is True


def test_result_roundtrip():
    d = 6
    buf = torch.zeros(d + HDR)
    g = torch.randn(d)
    res = WorkerResult(worker_id=3, g=g, ts=7, k_submit=55, nrows=1234,
                       elapsed_ms=8.5)
    pack_result(buf, d, res)
    out = unpack_result(buf, d, worker_id=3)
    assert torch.allclose(out.g, g)
    assert out.ts == 7 and out.k_submit == 55 and out.nrows == 1234
    assert abs(out.elapsed_ms - 8.5) < 1e-4
    assert out.worker_id == 3
