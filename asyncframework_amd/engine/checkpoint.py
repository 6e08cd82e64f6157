"""Checkpoint / resume of a run's optimizer state.

The reference has NO framework-level checkpointing (optimizer
state lives in driver memory only; its pitch is that SAGA's history replaces
lineage-based recovery). The MI355X rebuild adds a cheap periodic snapshot of
``(w, alpha_bar, per-worker history tables, k, clock)`` — off the hot path:
tensors are cloned to CPU on the updater thread's interval, serialized with
torch.save."""

from __future__ import annotations

import os
import tempfile
from typing import Dict, List, Optional

import torch

from .server import Server
from .worker import Worker


def capture_state(server: Server, workers: Optional[List[Worker]] = None,
                  remote_alpha: Optional[Dict[int, torch.Tensor]] = None
                  ) -> Dict:
    """Snapshot everything needed to resume (host-resident tensors).
    ``remote_alpha`` carries history tables gathered from remote ranks by
    the dist engine's snap sideband (engine/dist.py) keyed by worker id."""
    state = {
        "k": server.k,
        "current_time": server.AC.getCurrentTime(),
        "w": server.w.detach().cpu().clone(),
        "alpha_bar": (server.alpha_bar.detach().cpu().clone()
                      if server.alpha_bar is not None else None),
        "cfg": server.cfg.__dict__.copy(),
        "alpha": {},
    }
    for wk in workers or []:
        if wk.alpha is not None:
            state["alpha"][wk.id] = wk.alpha.detach().cpu().clone()
    for wid, t in (remote_alpha or {}).items():
        state["alpha"][wid] = t.detach().cpu().clone()
    return state


def save_state(path: str, state: Dict) -> None:
    """Atomic write (tmp + rename) so a crash mid-save keeps the previous
    checkpoint valid. ``state`` uses the capture_state schema — all engines
    (threads, dist, native dist) write interchangeable checkpoints."""
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".ckpt.tmp")
    os.close(fd)
    try:
        torch.save(state, tmp)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)


def save_checkpoint(path: str, server: Server,
                    workers: Optional[List[Worker]] = None,
                    remote_alpha: Optional[Dict[int, torch.Tensor]] = None
                    ) -> None:
    save_state(path, capture_state(server, workers, remote_alpha))


def load_checkpoint(path: str) -> Dict:
    # weights_only=True: the schema is tensors + primitives only, so there
    # is no reason to allow arbitrary unpickling (a hostile checkpoint file
    # could otherwise execute code at load).
    return torch.load(path, map_location="cpu", weights_only=True)


def restore(server: Server, workers: Optional[List[Worker]], state: Dict
            ) -> None:
    """Load a snapshot back into a server + workers (devices preserved).
    The worker-pool shape must match: SAGA history tables are keyed by
    worker id over a fixed sharding, so resuming under a different
    ``num_workers`` would silently mis-assign history."""
    ck_P = state.get("cfg", {}).get("num_workers")
    if ck_P is not None and ck_P != server.cfg.num_workers:
        raise ValueError(
            f"checkpoint was taken with num_workers={ck_P}, cannot resume "
            f"with num_workers={server.cfg.num_workers}")
    if state.get("cfg", {}).get("d") not in (None, server.cfg.d):
        raise ValueError("checkpoint dimensionality mismatch")
    server.k = int(state["k"])
    server.AC.setCurrentTime(int(state["current_time"]))
    server.w.copy_(state["w"].to(server.w.device))
    if state.get("alpha_bar") is not None and server.alpha_bar is not None:
        server.alpha_bar.copy_(state["alpha_bar"].to(server.alpha_bar.device))
    missing = []
    for wk in workers or []:
        if wk.alpha is None:
            continue
        if wk.id in state["alpha"]:
            wk.alpha.copy_(state["alpha"][wk.id].to(wk.alpha.device))
        else:
            missing.append(wk.id)
    if missing:
        # a checkpoint taken while a remote peer's snap sideband timed out
        # has no table for that worker; resuming silently with zeroed SAGA
        # history would corrupt the run's semantics — be loud about it.
        import warnings
        warnings.warn(
            f"checkpoint has no SAGA history table for worker(s) {missing}; "
            "their alpha stays as-is (zeros on a fresh engine)",
            RuntimeWarning, stacklevel=2)
