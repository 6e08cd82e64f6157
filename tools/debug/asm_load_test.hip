#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
struct B16x4 { union { ushort4 s; uint2 u; }; };
__global__ void k(const ushort* X, long total_elems, float* out, int d) {
  // emulate: row = blockIdx.x, lanes load (lane + it*64)*8 bytes
  const int lane = threadIdx.x & 63;
  int rr = __builtin_amdgcn_readfirstlane((int)blockIdx.x);
  const size_t off = (size_t)rr * d;
  const uint64_t rem = (uint64_t)(total_elems - off) * 2;
  const uint32_t nrec = rem > 0xFFFFFFF0ull ? 0xFFFFFFF0u : (uint32_t)rem;
  auto rs = __builtin_amdgcn_make_buffer_rsrc((void*)(X + off), 0, nrec, 0x00020000);
  B16x4 b[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    asm volatile("buffer_load_dwordx2 %0, %1, %2, 0 offen"
                 : "=v"(b[it].s) : "v"((lane + it * 64) * 8), "s"(rs));
  }
  asm volatile("s_waitcnt vmcnt(%[v])" ::[v] "i"(0));
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    asm volatile("" : "+v"(b[it].u.x));
    asm volatile("" : "+v"(b[it].u.y));
  }
  float z = 0;
#pragma unroll
  for (int it = 0; it < 4; ++it)
    z += (float)b[it].s.x + b[it].s.y + b[it].s.z + b[it].s.w;
  if (lane == 0) out[blockIdx.x] = z;
}
int main() {
  int n_rows = 100, d = 784;
  ushort* X; float* out;
  hipMalloc(&X, (size_t)n_rows * d * 2);
  hipMemset(X, 1, (size_t)n_rows * d * 2);
  hipMalloc(&out, n_rows * 4);
  hipLaunchKernelGGL(k, dim3(n_rows), dim3(64), 0, 0, X, (long)n_rows * d, out, d);
  hipError_t e = hipDeviceSynchronize();
  printf("err=%s\n", hipGetErrorString(e));
  float h[4]; hipMemcpy(h, out, 16, hipMemcpyDeviceToHost);
  printf("out0=%f out1=%f\n", h[0], h[1]);
  return 0;
}
