"""Engine configuration — the reference's 13 positional CLI knobs plus the
MI355X engine flags."""

from __future__ import annotations

from dataclasses import dataclass

import torch

_DTYPES = {
    "fp32": torch.float32,
    "fp64": torch.float64,
    "bf16": torch.bfloat16,
    "fp16": torch.float16,
}


@dataclass
class EngineConfig:
    # reference positional args (SparkASGDThread.scala:39-51)
    d: int = 0                      # num columns
    N: int = 0                      # num rows (whole dataset)
    num_workers: int = 8            # numPart
    num_iterations: int = 1000
    gamma: float = 0.01             # step size
    taw: int = 2 ** 30              # staleness bound tau
    batch_rate: float = 0.1         # b
    bucket_ratio: float = 0.93      # beta
    printer_freq: int = 100
    delay_coeff: float = -1.0       # coeff; -1 = cloud long-tail model
    seed: int = 42

    # engine flags (new)
    objective: str = "lsq"          # 'lsq' | 'logistic'
    algo: str = "asgd"              # 'asgd' | 'asaga'
    sync: bool = False
    dtype: str = "fp32"             # compute/storage dtype of X
    device: str = "cpu"
    history_placement: str = "device"  # 'device' (HBM) | 'host' (pinned DRAM)
    calib_factor: int = 100         # delay calibration window = calib_factor*P
    snapshot_weights: bool = True   # record optVars (time, w) for loss curves
    checkpoint_path: str = ""       # periodic optimizer-state snapshots
    checkpoint_every: int = 0       # every N applied updates (0 = off)
    worker_timeout_s: float = 0.0   # busy-worker timeout -> declared dead
                                    # (reference has none: a lost task left a
                                    # worker busy forever)

    def torch_dtype(self) -> torch.dtype:
        return _DTYPES[self.dtype]

    @property
    def par_recs(self) -> float:
        """Expected records per task: b*N/numPart
        (reference SparkASGDThread.scala:188)."""
        return self.batch_rate * self.N / self.num_workers

    @property
    def gate(self) -> int:
        """Worker quorum: floor(numPart*bucketRatio)
        (reference SparkASGDThread.scala:233-237)."""
        import math
        return int(math.floor(self.num_workers * self.bucket_ratio))
