"""Python adapter for the native C++ event-loop engine
(csrc/engine_native.cpp): multi-worker bounded-staleness async on one GPU
with zero Python in the round loop.

Each worker owns a shard, a weight-snapshot buffer (the
versioned-broadcast semantic), a gradient accumulator and a pinned-host
completion flag; the C++ loop does wave dispatch (one kernel per quorum
wave) -> pinned-flag poll -> tau filter -> batched fused update ->
quorum-gated redispatch, including the reference's straggler model and
calibration (csrc/engine_native.cpp header for the measured evolution).
Use for GPU multi-worker configs (the threaded Python engine stays for
CPU tests and as the semantics oracle)."""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from .config import EngineConfig
from .worker import Shard

_OBJ = {"lsq": 0, "logistic": 1}
_ALGO = {"asgd": 0, "asaga": 1}


class NativeLocalEngine:
    def __init__(self, cfg: EngineConfig, shards: List[Shard],
                 device: torch.device):
        from .. import _hip_core
        self._core = _hip_core
        assert device.type == "cuda", "native engine is GPU-only"
        self.cfg = cfg
        self.device = device
        self.shards = shards
        d = cfg.d
        self.w = torch.zeros(d, dtype=torch.float32, device=device)
        self.alpha_bar = torch.zeros(d, dtype=torch.float32, device=device)
        self._bufs = []
        self._keep = []   # keep tensors alive across the native call
        self.alpha_tables: List[torch.Tensor] = []
        for sh in shards:
            wbuf = torch.zeros(d, dtype=torch.float32, device=device)
            g = torch.zeros(d, dtype=torch.float32, device=device)
            ctr = torch.zeros(2, dtype=torch.int32, device=device)
            wd = dict(sparse=sh.is_sparse, y=sh.y.data_ptr(),
                      wbuf=wbuf.data_ptr(), g=g.data_ptr(),
                      ctr=ctr.data_ptr(), n_rows=sh.n_rows,
                      row_start=sh.row_start, x_is_bf16=0)
            keep = [wbuf, g, ctr]
            if sh.is_sparse:
                wd.update(indptr=sh.indptr.data_ptr(),
                          indices=sh.indices.data_ptr(),
                          values=sh.values.data_ptr())
                wd["x_is_bf16"] = 1 if sh.values.dtype == torch.bfloat16 else 0
            else:
                wd.update(X=sh.X.data_ptr())
                wd["x_is_bf16"] = 1 if sh.X.dtype == torch.bfloat16 else 0
            if cfg.algo == "asaga":
                cap = sh.n_rows if cfg.batch_rate >= 1.0 else min(
                    sh.n_rows, int(cfg.batch_rate * sh.n_rows * 3) + 4096)
                idx_out = torch.zeros(cap, dtype=torch.int32, device=device)
                e_out = torch.zeros(cap, dtype=torch.float32, device=device)
                if cfg.history_placement == "host":
                    # spill mode (BASELINE config 5): master table pinned in
                    # host DRAM; the device table becomes staging that the
                    # C++ loop refreshes per round via scan_rows +
                    # alpha_gather (mask-keyed, ~rate of the table)
                    master = torch.zeros(sh.n_rows, dtype=torch.float32,
                                         device="cpu").pin_memory()
                    alpha = torch.zeros(sh.n_rows, dtype=torch.float32,
                                        device=device)
                    srows = torch.empty(cap, dtype=torch.int32,
                                        device=device)
                    sylist = torch.empty(cap, dtype=torch.float32,
                                         device=device)
                    scnt = torch.zeros(1, dtype=torch.int32, device=device)
                    wd.update(alpha_host=master.data_ptr(),
                              srows=srows.data_ptr(),
                              sylist=sylist.data_ptr(),
                              scnt=scnt.data_ptr())
                    keep += [master, srows, sylist, scnt]
                    self.alpha_tables.append(master)
                else:
                    alpha = torch.zeros(sh.n_rows, dtype=torch.float32,
                                        device=device)
                    self.alpha_tables.append(alpha)
                wd.update(alpha=alpha.data_ptr(), idx_out=idx_out.data_ptr(),
                          e_out=e_out.data_ptr(), saga_cap=cap)
                keep += [alpha, idx_out, e_out]
            self._keep.extend(keep)
            self._bufs.append(wd)

    def run(self, num_iterations: Optional[int] = None,
            mark_lo: int = -1, mark_hi: int = -1,
            max_wall_s: float = 1800.0,
            snapshot_every: int = 0) -> Dict:
        """snapshot_every > 0 records (ms, w) optVars every that many
        applied updates (the reference's printer_freq loss-curve mechanism)
        into a device ring, returned as res['opt_vars']."""
        cfg = self.cfg
        iters = num_iterations or cfg.num_iterations
        snap_ring = None
        snap_cap = 0
        if snapshot_every > 0:
            snap_cap = iters // snapshot_every + 2
            snap_ring = torch.zeros(snap_cap, cfg.d, dtype=torch.float32,
                                    device=self.device)
        conf = dict(N=cfg.N, d=cfg.d, P=cfg.num_workers, iters=iters,
                    gamma=cfg.gamma, rate=cfg.batch_rate,
                    bucket_ratio=cfg.bucket_ratio, taw=cfg.taw,
                    seed=cfg.seed, algo=_ALGO[cfg.algo],
                    objective=_OBJ[cfg.objective], coeff=cfg.delay_coeff,
                    calib_window=cfg.calib_factor * cfg.num_workers,
                    mark_lo=mark_lo, mark_hi=mark_hi,
                    max_wall_s=max_wall_s, snap_every=snapshot_every,
                    snap_ring=snap_ring.data_ptr() if snap_ring is not None
                    else 0, snap_cap=snap_cap)
        torch.cuda.synchronize()
        res = self._core.native_local_run(conf, self._bufs,
                                          self.w.data_ptr(),
                                          self.alpha_bar.data_ptr())
        torch.cuda.synchronize()
        if snapshot_every > 0:
            snaps = res["snap_ms"]
            res["opt_vars"] = [(0, torch.zeros(cfg.d))] + [
                (int(ms), snap_ring[i].cpu().clone())
                for i, ms in enumerate(snaps)]
        return res

    def bench(self, warmup: int, steps: int,
              max_wall_s: float = 1800.0,
              snapshot_every: int = 0) -> Tuple[float, Dict]:
        """Exactly-K-steps contract: the native loop stamps wall times when
        k crosses warmup and warmup+steps (after a server-stream sync).
        snapshot_every > 0 also records the optVars loss curve (a 3 KB D2D
        copy on the server stream every that many applied updates)."""
        res = self.run(num_iterations=warmup + steps + 1,
                       mark_lo=warmup, mark_hi=warmup + steps,
                       max_wall_s=max_wall_s, snapshot_every=snapshot_every)
        t0, t1 = res["mark_lo_t"], res["mark_hi_t"]
        if not (t1 > t0 > 0):
            raise RuntimeError(f"native bench marks missing: {res}")
        return t1 - t0, res
