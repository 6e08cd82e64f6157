#!/usr/bin/env python3
"""Microbenchmark the gradient kernel in isolation (hipEvent timing).

Usage: python tools/prof_kernel.py [--rows R] [--cols D] [--iters N]
       [--rates 0.0,0.01] [--grids 512,1024,2048] [--pmc-mode]

--pmc-mode runs ONE config with few iters (for rocprofv3 --pmc wrapping).
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from asyncframework_amd import _hip_core  # noqa: E402


def time_kernel(X, y, w, g, g_part, n_ctr, rate, iters, use_part=True):
    stream = torch.cuda.current_stream().cuda_stream
    n_rows, d = X.shape
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    # warm
    for _ in range(3):
        _hip_core.grad_dense(X.data_ptr(), y.data_ptr(), w.data_ptr(),
                             g.data_ptr(),
                             g_part.data_ptr() if use_part else 0,
                             n_ctr.data_ptr(), 0, n_rows, d, 42, 1, 0, rate,
                             0, 1 if X.dtype == torch.bfloat16 else 0, stream)
    torch.cuda.synchronize()
    start.record()
    for i in range(iters):
        _hip_core.grad_dense(X.data_ptr(), y.data_ptr(), w.data_ptr(),
                             g.data_ptr(),
                             g_part.data_ptr() if use_part else 0,
                             n_ctr.data_ptr(), 0, n_rows, d, 42, i + 1, 0,
                             rate, 0,
                             1 if X.dtype == torch.bfloat16 else 0, stream)
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters  # ms


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=8_100_000)
    p.add_argument("--cols", type=int, default=784)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--rates", default="0.0,0.002,0.01")
    p.add_argument("--grids", default="0")
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--pmc-mode", action="store_true")
    p.add_argument("--csr", action="store_true",
                   help="rcv1-shape CSR kernel instead of dense")
    args = p.parse_args()
    if args.csr:
        csr_main(args)
        return

    dt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    dev = torch.device("cuda:0")
    gen = torch.Generator(device=dev).manual_seed(1)
    X = torch.randn(args.rows, args.cols, generator=gen, device=dev,
                    dtype=torch.float32).to(dt)
    y = torch.randn(args.rows, generator=gen, device=dev)
    w = torch.randn(args.cols, generator=gen, device=dev)
    g = torch.zeros(args.cols, device=dev)
    n_ctr = torch.zeros(1, dtype=torch.int32, device=dev)

    if args.pmc_mode:
        G = int(_hip_core.grad_grid(args.rows))
        g_part = torch.zeros(args.cols * G, device=dev)
        ms = time_kernel(X, y, w, g, g_part, n_ctr, 0.01, 5)
        print(f"pmc-mode done, {ms:.3f} ms/iter")
        return

    for gridstr in args.grids.split(","):
        grid = int(gridstr)
        if grid > 0:
            os.environ["ASYNCAMD_GRAD_GRID"] = str(grid)
        G = int(_hip_core.grad_grid(args.rows))
        g_part = torch.zeros(args.cols * G, device=dev)
        for rstr in args.rates.split(","):
            rate = float(rstr)
            ms = time_kernel(X, y, w, g, g_part, n_ctr, rate, args.iters)
            est_rows = int(rate * args.rows)
            bw = est_rows * args.cols * X.element_size() / (ms * 1e6)
            print(f"grid={G:5d} rate={rate:6.4f} rows~{est_rows:8d} "
                  f"-> {ms*1000:9.1f} us  ({bw:7.1f} GB/s payload)")


def csr_main(args):
    import numpy as np
    from asyncframework_amd.data.synthetic import synthetic_csr
    dev = torch.device("cuda:0")
    n, d = 697_641, 47_236
    indptr, indices, values, y = synthetic_csr(n, d, nnz_per_row=73, seed=1,
                                               device=dev)
    w = torch.randn(d, device=dev)
    g = torch.zeros(d, device=dev)
    n_ctr = torch.zeros(1, dtype=torch.int32, device=dev)
    stream = torch.cuda.current_stream().cuda_stream
    for gridstr in args.grids.split(","):
        if int(gridstr) > 0:
            os.environ["ASYNCAMD_GRAD_GRID"] = gridstr
        for rstr in args.rates.split(","):
            rate = float(rstr)
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            for _ in range(3):
                _hip_core.grad_csr(indptr.data_ptr(), indices.data_ptr(),
                                   values.data_ptr(), y.data_ptr(),
                                   w.data_ptr(), g.data_ptr(),
                                   n_ctr.data_ptr(), 0, n, 42, 1, 0, rate,
                                   0, 0, stream)
            torch.cuda.synchronize()
            start.record()
            for i in range(args.iters):
                _hip_core.grad_csr(indptr.data_ptr(), indices.data_ptr(),
                                   values.data_ptr(), y.data_ptr(),
                                   w.data_ptr(), g.data_ptr(),
                                   n_ctr.data_ptr(), 0, n, 42, i + 1, 0,
                                   rate, 0, 0, stream)
            end.record()
            torch.cuda.synchronize()
            ms = start.elapsed_time(end) / args.iters
            est = int(rate * n)
            bw = est * 73 * 6 / (ms * 1e6)
            print(f"csr grid={os.environ.get('ASYNCAMD_GRAD_GRID','auto'):>5s} "
                  f"rate={rate:6.4f} rows~{est:7d} -> {ms*1000:9.1f} us "
                  f"({bw:6.1f} GB/s nnz payload)")


if __name__ == "__main__":
    main()
