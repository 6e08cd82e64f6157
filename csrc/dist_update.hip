// Fused update kernels for the C++ dist server (csrc/server_dist.cpp).
// Same arithmetic as the local-engine kernels in csrc/kernels.hip
// (sgd_update_kernel / saga_update_kernel) — duplicated here because
// _dist_core.so is a torch extension built separately from _hip_core.so
// (ROADMAP §1: replace the aten add_ chain, 1-3 dispatcher launches per
// update on the default stream, with one HIP launch on the server's
// stream).

#include <hip/hip_runtime.h>

namespace {

__global__ void dist_sgd_update_kernel(float* __restrict__ w,
                                       const float* __restrict__ g,
                                       float gamma_k, float inv_batch,
                                       int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) w[i] -= gamma_k * inv_batch * g[i];
}

// w -= gamma*(inv_batch*g + alpha_bar_old); alpha_bar += inv_N*g — w reads
// the OLD alpha_bar (reference SparkASAGAThread.scala:217-220)
__global__ void dist_saga_update_kernel(float* __restrict__ w,
                                        const float* __restrict__ g,
                                        float* __restrict__ alpha_bar,
                                        float gamma, float inv_batch,
                                        float inv_N, int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) {
    const float gi = g[i];
    w[i] -= gamma * (inv_batch * gi + alpha_bar[i]);
    alpha_bar[i] += inv_N * gi;
  }
}

}  // namespace

extern "C" {

void launch_dist_sgd_update(float* w, const float* g, float gamma_k,
                            float inv_batch, int d, hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(dist_sgd_update_kernel, dim3(grid), dim3(256), 0,
                     stream, w, g, gamma_k, inv_batch, d);
}

void launch_dist_saga_update(float* w, const float* g, float* alpha_bar,
                             float gamma, float inv_batch, float inv_N,
                             int d, hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(dist_saga_update_kernel, dim3(grid), dim3(256), 0,
                     stream, w, g, alpha_bar, gamma, inv_batch, inv_N, d);
}

}  // extern "C"
