import os, sys, torch
sys.path.insert(0, os.getcwd())
from asyncframework_amd import ops
from asyncframework_amd.ops import torch_ref
from asyncframework_amd.utils.philox import bernoulli_mask

n, d = 4096, 784
g0 = torch.Generator(device="cuda").manual_seed(1)
X = torch.randn(n, d, generator=g0, device="cuda")
y = torch.randn(n, generator=g0, device="cuda")
w = torch.randn(d, generator=g0, device="cuda")
for rate in (0.01, 0.3):
    g, cnt = ops.grad_dense(X, y, w, seed=42, round_k=7, row_start=0, rate=rate)
    mask = torch.from_numpy(bernoulli_mask(42, 7, 0, n, rate)).cuda()
    gr, cr = torch_ref.grad_dense(X, y, w, mask, "lsq")
    rel = float((g - gr).norm() / (gr.norm() + 1e-12))
    print(f"rate={rate} cnt={cnt} ref={cr} rel={rel:.6f}")
    print("  kern:", g[:4].tolist())
    print("  ref: ", gr[:4].tolist())
