"""Dispatch/result message types exchanged between server and workers.

In-process these are plain Python objects holding tensor references; the
distributed engine packs them into fixed-size flat tensors for RCCL/gloo
point-to-point transfer."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class Dispatch:
    """Server -> worker round descriptor. ``w`` is the weight snapshot the
    worker must compute against (the versioned-broadcast semantic,
    reference ASYNCbroadcast.scala:21-27); ``ts`` the logical submit clock
    (reference RDD.scala:1100); ``k_submit`` the global iteration at submit
    (sampling seed round, reference ``sample(false,b,seed+k+1)``,
    SparkASGDThread.scala:314); ``accept_prev`` commits the worker's staged
    SAGA history from its previous round; ``delay_s`` the injected straggler
    delay; ``stop`` ends the worker loop; ``snap`` is a history-table
    sideband op for the dist engine (1 = worker SENDS its SAGA alpha table
    to the server for a checkpoint, 2 = worker RECEIVES a table to restore
    on resume) — no gradient round runs for snap messages."""
    w: Optional[torch.Tensor]
    ts: int = 0
    k_submit: int = 0
    accept_prev: bool = True
    delay_s: float = 0.0
    stop: bool = False
    snap: int = 0


@dataclass
class WorkerResult:
    """Worker -> server gradient envelope (pre-mailbox; the completion path
    wraps it into an RDDPartialRes)."""
    worker_id: int
    g: Optional[torch.Tensor]
    ts: int
    k_submit: int
    nrows: int
    elapsed_ms: float


# Header layout for the packed wire format (dist engine). The payload tensor
# is [d + HDR] floats: payload[:d] = w or g, payload[d:] = header.
HDR = 8
H_TS, H_K, H_ACCEPT, H_STOP, H_DELAY, H_NROWS, H_ELAPSED, H_SNAP = range(8)


def _hdr_tensor(vals) -> torch.Tensor:
    return torch.tensor(vals, dtype=torch.float32)


def pack_dispatch(buf: torch.Tensor, d: int, msg: Dispatch) -> None:
    # the header is float32 on the wire: integers are exact only to 2^24
    # (~16.7M updates — far above any BASELINE run length; fail loudly
    # rather than corrupt clocks silently if that is ever exceeded)
    assert msg.ts < (1 << 24) and msg.k_submit < (1 << 24), \
        "wire header exceeds float32 integer range"
    if msg.w is not None:
        buf[:d].copy_(msg.w.to(buf.dtype))
    # ONE host->device copy for the header (per-element writes into a GPU
    # tensor are one tiny H2D each — measured to dominate dispatch latency)
    hdr = _hdr_tensor([float(msg.ts), float(msg.k_submit),
                       1.0 if msg.accept_prev else 0.0,
                       1.0 if msg.stop else 0.0, msg.delay_s, 0.0, 0.0,
                       float(msg.snap)])
    buf[d:].copy_(hdr, non_blocking=False)


def unpack_dispatch(buf: torch.Tensor, d: int) -> Dispatch:
    h = buf[d:].tolist()
    return Dispatch(w=buf[:d], ts=int(h[H_TS]), k_submit=int(h[H_K]),
                    accept_prev=h[H_ACCEPT] > 0.5, stop=h[H_STOP] > 0.5,
                    delay_s=float(h[H_DELAY]), snap=int(h[H_SNAP]))


def pack_result(buf: torch.Tensor, d: int, res: WorkerResult) -> None:
    if res.g is not None:
        buf[:d].copy_(res.g.to(buf.dtype))
    hdr = _hdr_tensor([float(res.ts), float(res.k_submit), 0.0, 0.0, 0.0,
                       float(res.nrows), res.elapsed_ms, 0.0])
    buf[d:].copy_(hdr, non_blocking=False)


def unpack_result(buf: torch.Tensor, d: int, worker_id: int) -> WorkerResult:
    h = buf[d:].tolist()
    return WorkerResult(worker_id=worker_id, g=buf[:d], ts=int(h[H_TS]),
                        k_submit=int(h[H_K]), nrows=int(h[H_NROWS]),
                        elapsed_ms=float(h[H_ELAPSED]))
