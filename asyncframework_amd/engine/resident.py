"""Python adapter for the device-resident async engine
(csrc/engine_resident.hip): the ENTIRE bounded-staleness parameter-server
loop runs inside one persistent HIP kernel — a server block (single-writer
on w) plus G blocks per logical worker, communicating through device-scope
acquire/release atomics. Zero host API calls between launch and
completion; the host-driven native engine's ~6-12 us/update launch cost
disappears.

Dense ASGD/ASAGA only (CSR stays on the native engine). Co-residency is
checked at launch (a persistent kernel that oversubscribes the chip would
deadlock) and every device spin loop carries a realtime deadline, so a
wedged run aborts in seconds instead of hanging the GPU."""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from .config import EngineConfig
from .worker import Shard

_OBJ = {"lsq": 0, "logistic": 1}
_ALGO = {"asgd": 0, "asaga": 1}


class ResidentEngine:
    def __init__(self, cfg: EngineConfig, shards: List[Shard],
                 device: torch.device, blocks_per_worker: int = 0):
        from .. import _hip_core
        self._core = _hip_core
        assert device.type == "cuda", "resident engine is GPU-only"
        assert not shards[0].is_sparse, \
            "resident engine is dense-only (CSR -> native engine)"
        assert cfg.history_placement != "host", \
            "host-spill history -> native engine"
        assert 1 <= cfg.num_workers <= 64
        assert cfg.batch_rate < 1.0
        assert cfg.d <= 2048, \
            "resident engine caps d at 2048 (register-cached x chunks)"
        self.cfg = cfg
        self.device = device
        # G=0: auto-size to fill the chip (256 CUs want >>256 workgroups;
        # measured: G=31 at P=32 beats G=16 by ~8%). The launch retries
        # with halved G if occupancy rejects the grid (run() below).
        self.G = blocks_per_worker or max(2, min(31, 1008 // cfg.num_workers))
        d = cfg.d
        self.w = torch.zeros(d, dtype=torch.float32, device=device)
        self.alpha_bar = torch.zeros(d, dtype=torch.float32, device=device)
        self.alpha_tables: List[torch.Tensor] = []
        self._keep = []
        desc = torch.zeros(cfg.num_workers, 8, dtype=torch.int64)
        self._x_is_bf16 = 1 if shards[0].X.dtype == torch.bfloat16 else 0
        for i, sh in enumerate(shards):
            assert sh.X.is_contiguous() and sh.y.dtype == torch.float32
            wbuf = torch.zeros(d, dtype=torch.float32, device=device)
            g = torch.zeros(d, dtype=torch.float32, device=device)
            desc[i, 0] = sh.X.data_ptr()
            desc[i, 1] = sh.y.data_ptr()
            desc[i, 2] = wbuf.data_ptr()
            desc[i, 3] = g.data_ptr()
            desc[i, 4] = sh.n_rows
            desc[i, 5] = sh.row_start
            self._keep += [wbuf, g, sh.X, sh.y]
            if cfg.algo == "asaga":
                alpha = torch.zeros(sh.n_rows, dtype=torch.float32,
                                    device=device)
                stage = torch.zeros(sh.n_rows, dtype=torch.float32,
                                    device=device)
                desc[i, 6] = alpha.data_ptr()
                desc[i, 7] = stage.data_ptr()
                self.alpha_tables.append(alpha)
                self._keep += [alpha, stage]
        self._desc = desc.to(device)
        P = cfg.num_workers
        # one 128-B cache line PER WORKER per flag (RES_CSTRIDE in the
        # kernel): hundreds of spinning blocks poll these with atomic RMWs,
        # and packed flags serialize every poll on 1-2 coherent lines
        self._ctl = {
            name: torch.zeros(P * 32, dtype=torch.int32, device=device)
            for name in ("go_round", "go_key", "go_flags", "done_round",
                         "done_ctr")
        }
        # 26 counters + RES_MAXP packed per-worker abort-diagnostic slots
        # (OUT_N in csrc/engine_resident.hip; 96 leaves headroom)
        self._out = torch.zeros(96, dtype=torch.int64, device=device)

    def run(self, num_iterations: Optional[int] = None,
            mark_lo: int = -1, mark_hi: int = -1,
            max_wall_s: float = 600.0,
            snapshot_every: int = 0) -> Dict:
        cfg = self.cfg
        iters = num_iterations or cfg.num_iterations
        snap_cap = (iters // snapshot_every + 2) if snapshot_every > 0 else 0
        snap_ring = (torch.zeros(snap_cap, cfg.d, dtype=torch.float32,
                                 device=self.device)
                     if snap_cap else None)
        snap_cycles = (torch.zeros(snap_cap, dtype=torch.int64,
                                   device=self.device)
                       if snap_cap else None)
        for t in self._ctl.values():
            t.zero_()
        self._out.zero_()
        gate = max(1, int(cfg.num_workers * cfg.bucket_ratio))
        par_recs = cfg.batch_rate * cfg.N / cfg.num_workers
        conf = dict(
            w=self.w.data_ptr(), desc=self._desc.data_ptr(),
            alpha_bar=self.alpha_bar.data_ptr(),
            snap_ring=snap_ring.data_ptr() if snap_ring is not None else 0,
            snap_cycles=(snap_cycles.data_ptr()
                         if snap_cycles is not None else 0),
            out=self._out.data_ptr(),
            N=cfg.N, d=cfg.d, P=cfg.num_workers, G=self.G, iters=iters,
            gamma=cfg.gamma, rate=cfg.batch_rate, seed=cfg.seed,
            taw=cfg.taw, gate=gate, x_is_bf16=self._x_is_bf16,
            algo=_ALGO[cfg.algo], objective=_OBJ[cfg.objective],
            coeff=cfg.delay_coeff,
            calib_window=cfg.calib_factor * cfg.num_workers,
            mark_lo=mark_lo, mark_hi=mark_hi,
            snap_every=snapshot_every, snap_cap=snap_cap,
            inv_batch=1.0 / par_recs, inv_N=1.0 / cfg.N,
            max_wall_s=max_wall_s,
            **{k: v.data_ptr() for k, v in self._ctl.items()},
        )
        torch.cuda.synchronize()
        while True:
            try:
                res = self._core.resident_run(conf)
                break
            except RuntimeError as e:
                # co-residency refusal (grid would not be simultaneously
                # schedulable => persistent kernel would deadlock): halve G
                if "co-resident" in str(e) and self.G > 2:
                    self.G = max(2, self.G // 2)
                    conf["G"] = self.G
                    continue
                raise
        torch.cuda.synchronize()
        res["wall_exhausted"] = False
        if res["aborted"]:
            # A LIVE run that burned its wall budget (completions were still
            # being processed near the deadline — e.g. tau=0 rejecting
            # nearly everything) returns a partial result, exactly like the
            # host engines' max_wall_s path. Only a run with NO recent
            # completion progress is treated as a protocol wedge.
            gap_ms = res["elapsed_ms"] - res["last_progress_ms"]
            if res["last_progress_ms"] >= 0 and gap_ms < 500.0:
                res["wall_exhausted"] = True
            else:
                dump = self._out[26:26 + cfg.num_workers].cpu().tolist()
                state = [
                    dict(w=i, busy=v & 1, hold=(v >> 1) & 1,
                         round_no=(v >> 8) & 0xFFFFFFFF,
                         ksub=(v >> 40) & 0xFFFF,
                         go=int(self._ctl["go_round"][i * 32]),
                         done=int(self._ctl["done_round"][i * 32]),
                         ctr=int(self._ctl["done_ctr"][i * 32]))
                    for i, v in enumerate(dump)]
                raise RuntimeError(
                    f"resident engine wedged at k={res['k']} (device "
                    f"deadline {max_wall_s}s hit with no completion "
                    f"progress for {gap_ms:.0f}ms); per-worker state: "
                    f"{state}")
        if snapshot_every > 0:
            n = int(res["snap_n"])
            cpm = res["cycles_per_ms"]
            t0c = res["t0_cycles"]
            ring = snap_ring[:n].cpu()
            cyc = snap_cycles[:n].cpu()
            res["opt_vars"] = [(0, torch.zeros(cfg.d))] + [
                (int((int(cyc[i]) - t0c) / cpm), ring[i].clone())
                for i in range(n)]
        return res

    def bench(self, warmup: int, steps: int, max_wall_s: float = 600.0,
              snapshot_every: int = 0) -> Tuple[float, Dict]:
        res = self.run(num_iterations=warmup + steps + 1,
                       mark_lo=warmup, mark_hi=warmup + steps,
                       max_wall_s=max_wall_s, snapshot_every=snapshot_every)
        t0, t1 = res["mark_lo_t"], res["mark_hi_t"]
        if not (t1 > t0 > 0):
            raise RuntimeError(f"resident bench marks missing: {res}")
        return t1 - t0, res
