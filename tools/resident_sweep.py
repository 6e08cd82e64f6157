"""Resident-engine perf sweep on the flagship (mnist8m-shape) config.

Run on a GPU box:
    python tools/resident_sweep.py > gpurun_out/res_sweep.log 2>&1

Prints one line per config:
    <tag> ups=<updates/s> grad_ms/round=... spin_ms/round=...
    srv_sweep=<ms total> srv_disp=<ms total> el=<ms>
"""
import sys

import torch

sys.path.insert(0, ".")
from asyncframework_amd.data.shard import row_shards          # noqa: E402
from asyncframework_amd.data.synthetic import synthetic_dense  # noqa: E402
from asyncframework_amd.engine.config import EngineConfig      # noqa: E402
from asyncframework_amd.engine.resident import ResidentEngine  # noqa: E402
from asyncframework_amd.engine.worker import Shard             # noqa: E402


def run(P, rows, d, dtype, iters, G, tag, rate=0.01, algo="asgd"):
    dev = torch.device("cuda:0")
    X, y = synthetic_dense(rows, d, seed=42, dtype=dtype, device=dev)
    cfg = EngineConfig(
        d=d, N=rows, num_workers=P, num_iterations=iters, gamma=1.5625e-3,
        taw=20_000_000, batch_rate=rate, bucket_ratio=0.7,
        printer_freq=1 << 30, seed=42, algo=algo, objective="lsq",
        dtype="bf16" if dtype == torch.bfloat16 else "fp32",
        device="cuda:0")
    shards = [Shard(row_start=s, n_rows=t - s, X=X[s:t], y=y[s:t])
              for s, t in row_shards(rows, P)]
    eng = ResidentEngine(cfg, shards, dev, blocks_per_worker=G)
    res = eng.run(num_iterations=iters, max_wall_s=120.0)
    el = res["elapsed_ms"]
    k = res["k"]
    rounds = max(1, res["w0_rounds"])
    print(f"{tag} ups={k / el * 1000:.0f} "
          f"grad_ms/round={res['w0_grad_ms'] / rounds:.3f} "
          f"spin_ms/round={res['w0_spin_ms'] / rounds:.3f} "
          f"srv_sweep={res['srv_sweep_ms']:.0f} "
          f"srv_disp={res['srv_dispatch_ms']:.0f} el={el:.0f} "
          f"rej={res['rejected']} stale={res['max_staleness']} "
          f"clas={res['srv_classify_ms']:.1f} "
          f"apply={res['srv_apply_ms']:.1f} "
          f"wsw={res['srv_work_sweeps']} "
          f"wake_us={res['w0_wake_ms'] / max(1, res['w0_wake_n']) * 1e3:.1f} "
          f"det_us={res['w0_detect_ms'] / max(1, res['w0_detect_n']) * 1e3:.1f}",
          flush=True)
    del eng, X, y
    torch.cuda.empty_cache()


if __name__ == "__main__":
    # P=1 isolated: one worker's shard of the P=32 flagship split
    run(1, 253_125, 784, torch.bfloat16, 2000, G=31, tag="P1-iso-bf16-G31")
    run(32, 8_100_000, 784, torch.bfloat16, 20000, G=31, tag="P32-bf16-G31")
    run(32, 8_100_000, 784, torch.bfloat16, 20000, G=16, tag="P32-bf16-G16")
    run(32, 8_100_000, 784, torch.float32, 20000, G=31, tag="P32-fp32-G31")
    run(48, 8_100_000, 784, torch.bfloat16, 20000, G=21, tag="P48-bf16-G21")
    run(64, 8_100_000, 784, torch.bfloat16, 20000, G=15, tag="P64-bf16-G15")
