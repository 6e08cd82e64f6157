// Shared between kernels.hip (kernel + launcher) and engine_native.cpp
// (batch assembly): the per-batch argument block for the native engine's
// batched update kernel. Rides in the kernel-arg space (<= 4 KB), so a
// batch costs one launch and no H2D staging.
#pragma once

#define MU_MAX 32

struct MultiUpdateArgs {
  int n;                 // accepted gradients in this batch
  int m;                 // snapshot targets (workers redispatching now)
  int algo;              // 0 asgd, 1 asaga
  int gw[MU_MAX];        // worker id per gradient (indexes g_tab)
  int sw[MU_MAX];        // worker id per snapshot (indexes wbuf_tab)
  float scale[MU_MAX];   // ASGD: gamma_k * inv_batch per gradient
};
