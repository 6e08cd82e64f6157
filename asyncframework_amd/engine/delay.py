"""Straggler / delay injection — the fault-injection subsystem.

Faithful rebuild of the reference's in-app heterogeneity simulator
(SparkASGDThread.scala:124-141 straggler selection, :287-312 injection,
:177-186,:247-252 calibration):

* 25% of workers are stragglers: ``length = round(0.25*P)``; the first
  ``length - round(0.8*length)`` of them (indices ``c*4``) are long-tail
  (2.5-10x), the rest normal (1.5-2.5x).
* ``coeff == -1`` selects the cloud long-tail model; ``coeff > 0`` slows
  only worker 0 by ``coeff * avgDelay``.
* ``avgDelay`` is calibrated from the first ``100*P`` completed tasks
  (arrival - submit), activated once ``k > 100*P``.

The reference draws the per-round slowdown from an unseeded RNG; the rebuild
uses Philox keyed on (seed, round, worker) so runs are reproducible.
"""

from __future__ import annotations

import threading

from ..utils.philox import uniform01


class DelayInjector:
    def __init__(self, num_workers: int, coeff: float, seed: int = 42,
                 calib_window: int | None = None):
        self.P = num_workers
        self.coeff = coeff
        self.seed = seed
        self.cloud = coeff == -1.0
        self.calib_window = (calib_window if calib_window is not None
                             else 100 * num_workers)
        length = int(round(0.25 * num_workers))
        length_normal = int(round(0.8 * length))
        length_longtail = length - length_normal
        self.straggler_longtail = set()
        self.straggler_normal = set()
        for c in range(length):
            if c < length_longtail:
                self.straggler_longtail.add(c * 4)
            else:
                self.straggler_normal.add(c * 4)
        self._lock = threading.Lock()
        self._cul_time = 0.0
        self._cul_count = 0
        self.avg_delay_ms = 0.0
        self.flag = False  # set once k > calib_window

    def record_task(self, k: int, task_ms: float) -> None:
        """Calibration sample (reference updater thread,
        SparkASGDThread.scala:177-186): only while k < calib window."""
        if k < self.calib_window:
            with self._lock:
                self._cul_time += task_ms
                self._cul_count += 1

    def maybe_activate(self, k: int) -> None:
        """Reference main loop :247-252: first round with k > window fixes
        avgDelay = culTime/culCount and raises the flag."""
        if not self.flag and k > self.calib_window:
            with self._lock:
                if self._cul_count > 0:
                    self.avg_delay_ms = self._cul_time / self._cul_count
            self.flag = True

    def delay_ms(self, worker_id: int, round_k: int) -> float:
        """Injected sleep for this worker at this round, in ms."""
        if not self.flag or self.coeff == 0.0:
            return 0.0
        if not self.cloud:
            if worker_id == 0 and self.coeff > 0:
                return round(self.coeff * self.avg_delay_ms)
            return 0.0
        if worker_id in self.straggler_longtail:
            u = float(uniform01(self.seed, round_k, worker_id, 1)[0])
            return round((u * 7.5 + 2.5) * self.avg_delay_ms)
        if worker_id in self.straggler_normal:
            u = float(uniform01(self.seed, round_k, worker_id, 1)[0])
            return round((u + 1.5) * self.avg_delay_ms)
        return 0.0
