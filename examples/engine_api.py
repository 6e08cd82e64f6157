#!/usr/bin/env python3
"""The engine API (the hot path): configure, shard, run, report — the same
calls the CLI drivers and bench.py make.

    python examples/engine_api.py              # CPU threads engine
    # on a GPU box: swap device="cuda:0" (and engine="native" via run_engine)
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from asyncframework_amd.engine.config import EngineConfig
from asyncframework_amd.run import (build_dense_workers, final_report,
                                    load_dataset, run_engine)

cfg = EngineConfig(d=32, N=4000, num_workers=4, num_iterations=400,
                   gamma=0.3, taw=1 << 30, batch_rate=0.2, bucket_ratio=0.7,
                   printer_freq=100, delay_coeff=0.0, seed=42,
                   algo="asgd", objective="lsq", dtype="fp32", device="cpu")
data = load_dataset(cfg, "synthetic", "synthetic")
workers = build_dense_workers(cfg, *data)
res, _server = run_engine(cfg, workers, engine="threads")
final_report(cfg, res, data, sparse=False)
