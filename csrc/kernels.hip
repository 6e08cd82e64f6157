// Hand-written CDNA4 (gfx950 / MI355X) kernels for the ASYNC hot path.
//
// Kernel inventory (reference JVM hot loops they replace):
//   K1 grad_dense      — fused Philox sample mask + per-row dot + scaled
//                        accumulate (reference gradfun SparkASGDThread.scala:
//                        423-438 + reducePartition fold RDD.scala:1103-1123).
//   K2 grad_csr        — CSR SpMV-style gradient, wave-per-row
//                        (reference sparse BLAS.scala:74-90,134-160).
//   K3 saga_grad_*     — K1/K2 fused with the per-sample history gather and
//                        staged scalar emit (SparkASAGAThread.scala:380-385).
//   K5 sgd_update      — fused scale+axpy weight update (:188-192).
//   K6 saga_update     — fused SAGA triple-axpy (:217-220).
//   K8 philox.h        — in-kernel counter-based Bernoulli mask.
//
// Design notes (MI355X, measured on hardware):
//  * 64-wide wavefronts. ONE Philox eval decides FOUR consecutive rows
//    (counter = row/4, output word = row%4) — the full-dataset mask scan is
//    the dominant fixed cost per round (8.1M rows; 32-bit integer multiply
//    is slow on the VALU), so each wave covers 256 rows per eval step.
//  * row processing is latency-bound, not bandwidth-bound, at the
//    reference's sampling rates (b=0.01 samples ~81k of 8.1M rows): rows are
//    processed by SUB-WAVES of LPR lanes (template param 64/32/16), so a
//    wave keeps 64/LPR rows in flight — raising memory-level parallelism
//    without extra registers.
//  * dense path: w staged in LDS; per-sub-wave fp32 gradient slabs in LDS
//    (LDS = (1 + 4*64/LPR)*d*4 bytes); finalize writes per-block partials
//    to a transposed slab, summed by reduce_partials (no global atomics:
//    a d-wide atomicAdd finalize serialized on 784 addresses).
//  * sparse path: w stays in L2 (rcv1 d=47236 -> 189 KB; L2 is 4 MiB/XCD);
//    gradient scatter via global fp32 atomics (~73 nnz/row, low contention).
//  * bf16 rows load as ushort4 (8 B/lane) — scalar bf16 loads halve
//    effective bandwidth (CDNA guide, common mistake #2).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cstdlib>
#include "philox.h"

#define WAVE 64
#define BLOCK 256
#define WAVES_PER_BLOCK (BLOCK / WAVE)
#define ROWS_PER_WAVE 256  // 64 lanes x 4 rows per philox eval
#define ROWS_PER_BLOCK_ITER (WAVES_PER_BLOCK * ROWS_PER_WAVE)

// ---------------------------------------------------------------- helpers

__device__ __forceinline__ float link_residual(float z, float yv, int obj) {
  if (obj == 1) return 1.0f / (1.0f + __expf(-z)) - yv;  // logistic
  return z - yv;                                         // lsq
}

template <typename XT>
__device__ __forceinline__ float to_f32(XT v);
template <>
__device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

// Load 4 consecutive elements starting at xrow[4*j4] as floats.
template <typename XT>
__device__ __forceinline__ void load4(const XT* xrow, int j4, float out[4]);

template <>
__device__ __forceinline__ void load4<float>(const float* xrow, int j4,
                                             float out[4]) {
  const float4 v = reinterpret_cast<const float4*>(xrow)[j4];
  out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
}

template <>
__device__ __forceinline__ void load4<__hip_bfloat16>(
    const __hip_bfloat16* xrow, int j4, float out[4]) {
  const ushort4 v = reinterpret_cast<const ushort4*>(xrow)[j4];
  union { unsigned short u; __hip_bfloat16 b; } c0{v.x}, c1{v.y}, c2{v.z},
      c3{v.w};
  out[0] = __bfloat162float(c0.b); out[1] = __bfloat162float(c1.b);
  out[2] = __bfloat162float(c2.b); out[3] = __bfloat162float(c3.b);
}

// ---------------------------------------------------------------- K1 (+K3)

// LPR = lanes per row (64, 32 or 16): rows are processed by aligned
// sub-waves so 64/LPR rows are in flight per wave.
template <typename XT, bool SAGA, int LPR>
__global__ __launch_bounds__(BLOCK) void grad_dense_kernel(
    const XT* __restrict__ X, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    float* __restrict__ g_part, int* __restrict__ n_out,
    float* __restrict__ alpha, int* __restrict__ idx_out,
    float* __restrict__ e_out, int* __restrict__ pos_ctr,
    const int* __restrict__ k_dev, int commit_now, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, uint32_t threshold,
    int take_all, int objective) {
  if (k_dev) round_k = (uint32_t)(*k_dev) + 1u;  // graph mode: round = k+1
  constexpr int NSUB = WAVE / LPR;
  constexpr int NSLAB = WAVES_PER_BLOCK * NSUB;
  extern __shared__ float smem[];
  float* w_lds = smem;           // [d]
  float* gacc = smem + d;        // [NSLAB][d]
  for (int j = threadIdx.x; j < d; j += BLOCK) {
    w_lds[j] = w[j];
#pragma unroll
    for (int s2 = 0; s2 < NSLAB; ++s2) gacc[(size_t)s2 * d + j] = 0.f;
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int sub = lane / LPR;
  const int sl = lane % LPR;
  float* gw = gacc + (size_t)(wave * NSUB + sub) * d;
  int local_count = 0;
  const int d4 = d >> 2;

  const long gstride = (long)gridDim.x * ROWS_PER_BLOCK_ITER;
  for (long bb = (long)blockIdx.x * ROWS_PER_BLOCK_ITER; bb < n_rows;
       bb += gstride) {
    const long base = bb + (long)wave * ROWS_PER_WAVE;
    if (base >= n_rows) continue;
    // scan 256 rows: lane's philox block covers rows base+4*lane..+3
    // (requires row_start % 4 == 0 — enforced by the launcher/sharder)
    const uint4 x = philox_block4(
        seed, round_k, (row_start + (uint64_t)base) / 4 + (uint64_t)lane);
    const long rem = n_rows - base;
    unsigned long long m[4];
    {
      const long lrow = 4L * lane;
      m[0] = __ballot(lrow + 0 < rem && (take_all || x.x < threshold));
      m[1] = __ballot(lrow + 1 < rem && (take_all || x.y < threshold));
      m[2] = __ballot(lrow + 2 < rem && (take_all || x.z < threshold));
      m[3] = __ballot(lrow + 3 < rem && (take_all || x.w < threshold));
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      unsigned long long mm = m[i];
      for (int s2 = 0; s2 < sub && mm; ++s2) mm &= mm - 1;  // my first bit
      while (mm) {
        const int bit = __ffsll((long long)mm) - 1;
        const long rr = base + 4L * bit + i;
        const XT* xrow = X + (size_t)rr * d;
        float z = 0.f;
        for (int j4 = sl; j4 < d4; j4 += LPR) {
          float xv[4];
          load4<XT>(xrow, j4, xv);
          const int j = j4 * 4;
          z += xv[0] * w_lds[j] + xv[1] * w_lds[j + 1] +
               xv[2] * w_lds[j + 2] + xv[3] * w_lds[j + 3];
        }
        for (int j = d4 * 4 + sl; j < d; j += LPR)  // d % 4 tail
          z += to_f32<XT>(xrow[j]) * w_lds[j];
#pragma unroll
        for (int off = LPR / 2; off > 0; off >>= 1)
          z += __shfl_xor(z, off, WAVE);
        float e = link_residual(z, y[rr], objective);
        float coeff = e;
        if (SAGA) {
          const float a_old = alpha[rr];
          coeff = e - a_old;
          if (sl == 0) {
            if (commit_now) {
              // sequential graph mode: every round accepted -> commit in
              // place (each row sampled once per round, read-before-write)
              alpha[rr] = e;
            } else {
              const int pos = atomicAdd(pos_ctr, 1);
              idx_out[pos] = (int)rr;
              e_out[pos] = e;
            }
          }
        }
        ++local_count;
        for (int j4 = sl; j4 < d4; j4 += LPR) {
          float xv[4];
          load4<XT>(xrow, j4, xv);
          const int j = j4 * 4;
          gw[j] += coeff * xv[0]; gw[j + 1] += coeff * xv[1];
          gw[j + 2] += coeff * xv[2]; gw[j + 3] += coeff * xv[3];
        }
        for (int j = d4 * 4 + sl; j < d; j += LPR)  // d % 4 tail
          gw[j] += coeff * to_f32<XT>(xrow[j]);
        for (int s2 = 0; s2 < NSUB && mm; ++s2) mm &= mm - 1;  // next mine
      }
    }
  }
  __syncthreads();
  if (g_part != nullptr) {
    // plain-store partials g_part[j*G + b]; reduce_partials sums them
    const size_t G = gridDim.x;
    for (int j = threadIdx.x; j < d; j += BLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NSLAB; ++s2) s += gacc[(size_t)s2 * d + j];
      g_part[(size_t)j * G + blockIdx.x] = s;
    }
  } else {
    for (int j = threadIdx.x; j < d; j += BLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NSLAB; ++s2) s += gacc[(size_t)s2 * d + j];
      if (s != 0.f) atomicAdd(&g_out[j], s);
    }
  }
  if (sl == 0 && local_count) atomicAdd(n_out, local_count);
}

// ------------------------------------------------- K1 pipelined (queue)
//
// Two-phase dense gradient: (1) Philox scan appends sampled local row ids to
// an LDS queue (block-shared, atomic append; the near-full guard makes
// overflow impossible), (2) waves drain the queue round-robin with a
// DEPTH-deep software pipeline (statically unrolled — runtime-indexed
// register arrays go to scratch, CDNA guide rule 20): while row q computes,
// DEPTH-1 later rows' loads are in flight, and the accumulate pass reuses
// the registers instead of re-reading X. This is the latency fix: at b=0.01
// a wave sees ~1 sampled row per 256-row scan group, so the non-queued
// kernel paid a full HBM latency per row (measured ~7 us serial per row).

// queue capacity must exceed rows-per-scan-iteration (the near-full guard
// is `qn >= QCAP - RPB`; QCAP == RPB would never scan and livelock)
#define QCAP_FOR(PB) ((PB) == 512 ? 4096 : 2048)
#define PIPE_MAXIT 8  // supports d <= 2048, d % 4 == 0

template <typename XT> struct RowVec;
template <> struct RowVec<float> { using T = float4; };
template <> struct RowVec<__hip_bfloat16> { using T = ushort4; };

__device__ __forceinline__ void cvt4(const float4& r, float o[4]) {
  o[0] = r.x; o[1] = r.y; o[2] = r.z; o[3] = r.w;
}
__device__ __forceinline__ void cvt4(const ushort4& r, float o[4]) {
  union { unsigned short u; __hip_bfloat16 b; } c0{r.x}, c1{r.y}, c2{r.z},
      c3{r.w};
  o[0] = __bfloat162float(c0.b); o[1] = __bfloat162float(c1.b);
  o[2] = __bfloat162float(c2.b); o[3] = __bfloat162float(c3.b);
}

template <typename XT>
__device__ __forceinline__ void load_row_regs(
    const XT* __restrict__ xrow, int lane, int d4,
    typename RowVec<XT>::T raw[PIPE_MAXIT]) {
#pragma unroll
  for (int it = 0; it < PIPE_MAXIT; ++it) {
    const int j4 = lane + it * WAVE;
    if (j4 < d4)
      raw[it] = reinterpret_cast<const typename RowVec<XT>::T*>(xrow)[j4];
  }
}

template <typename XT, bool SAGA, int PBLOCK, int DEPTH>
__global__ __launch_bounds__(PBLOCK) void grad_dense_pipe_kernel(
    const XT* __restrict__ X, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    float* __restrict__ g_part, int* __restrict__ n_out,
    float* __restrict__ alpha, int* __restrict__ idx_out,
    float* __restrict__ e_out, int* __restrict__ pos_ctr,
    const int* __restrict__ k_dev, int commit_now, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, uint32_t threshold,
    int take_all, int objective) {
  if (k_dev) round_k = (uint32_t)(*k_dev) + 1u;
  constexpr int NW = PBLOCK / WAVE;            // waves per block
  constexpr int RPB = NW * ROWS_PER_WAVE;      // rows scanned per block-iter
  extern __shared__ float smem[];
  float* w_lds = smem;                          // [d]
  float* gacc = smem + d;                       // [NW][d]
  constexpr int QCAP = QCAP_FOR(PBLOCK);
  static_assert(QCAP >= RPB + 1024, "queue must out-size one scan iter");
  int* rowq = (int*)(smem + (size_t)(1 + NW) * d);  // [QCAP]
  int* qn = rowq + QCAP;
  for (int j = threadIdx.x; j < d; j += PBLOCK) {
    w_lds[j] = w[j];
#pragma unroll
    for (int s2 = 0; s2 < NW; ++s2) gacc[(size_t)s2 * d + j] = 0.f;
  }
  if (threadIdx.x == 0) *qn = 0;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float* gw = gacc + (size_t)wave * d;
  int local_count = 0;
  const int d4 = d >> 2;
  using RV = typename RowVec<XT>::T;

  const long ngroups = (n_rows + RPB - 1) / RPB;
  long gi = blockIdx.x;
  bool done = false;
  while (!done) {
    // ---- scan phase: fill the queue until near-full or rows exhausted
    while (true) {
      __syncthreads();
      if (gi >= ngroups || *qn >= QCAP - RPB) break;
      const long base = gi * (long)RPB + (long)wave * ROWS_PER_WAVE;
      if (base < n_rows) {
        const uint4 x = philox_block4(
            seed, round_k, (row_start + (uint64_t)base) / 4 + (uint64_t)lane);
        const long rem = n_rows - base;
        const long lrow = base + 4L * lane;
        const uint32_t xs[4] = {x.x, x.y, x.z, x.w};
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (4L * lane + i < rem && (take_all || xs[i] < threshold)) {
            const int pos = atomicAdd(qn, 1);
            rowq[pos] = (int)(lrow + i);
          }
        }
      }
      gi += gridDim.x;
    }
    __syncthreads();
    const int nq = min(*qn, QCAP);
    // ---- process phase: wave-strided, DEPTH-deep pipeline (static bufs)
    RV buf[DEPTH][PIPE_MAXIT];
#pragma unroll
    for (int p = 0; p < DEPTH; ++p) {
      const int q = wave + p * NW;
      if (q < nq) load_row_regs<XT>(X + (size_t)rowq[q] * d, lane, d4, buf[p]);
    }
    int q_base = wave;
    while (q_base < nq) {
#pragma unroll
      for (int p = 0; p < DEPTH; ++p) {
        const int q = q_base + p * NW;
        if (q < nq) {
          const int rr = rowq[q];
          float z = 0.f;
#pragma unroll
          for (int it = 0; it < PIPE_MAXIT; ++it) {
            const int j4 = lane + it * WAVE;
            if (j4 < d4) {
              float o[4];
              cvt4(buf[p][it], o);
              // b128 LDS read: 16-lane groups hit distinct banks (b32 had
              // lanes l and l+8 colliding -> 26% LDSBankConflict measured)
              const float4 wv = reinterpret_cast<const float4*>(w_lds)[j4];
              z += o[0] * wv.x + o[1] * wv.y + o[2] * wv.z + o[3] * wv.w;
            }
          }
#pragma unroll
          for (int off = 32; off > 0; off >>= 1)
            z += __shfl_xor(z, off, WAVE);
          float e = link_residual(z, y[rr], objective);
          float coeff = e;
          if (SAGA) {
            const float a_old = alpha[rr];
            coeff = e - a_old;
            if (lane == 0) {
              if (commit_now) {
                alpha[rr] = e;
              } else {
                const int pos = atomicAdd(pos_ctr, 1);
                idx_out[pos] = rr;
                e_out[pos] = e;
              }
            }
          }
          ++local_count;
#pragma unroll
          for (int it = 0; it < PIPE_MAXIT; ++it) {
            const int j4 = lane + it * WAVE;
            if (j4 < d4) {
              float o[4];
              cvt4(buf[p][it], o);
              float4* gw4 = reinterpret_cast<float4*>(gw);
              float4 cur = gw4[j4];
              cur.x += coeff * o[0]; cur.y += coeff * o[1];
              cur.z += coeff * o[2]; cur.w += coeff * o[3];
              gw4[j4] = cur;
            }
          }
          // refill this buffer DEPTH rows ahead
          const int qf = q + DEPTH * NW;
          if (qf < nq)
            load_row_regs<XT>(X + (size_t)rowq[qf] * d, lane, d4, buf[p]);
        }
      }
      q_base += DEPTH * NW;
    }
    __syncthreads();
    if (threadIdx.x == 0) *qn = 0;
    done = gi >= ngroups;
  }
  __syncthreads();
  if (g_part != nullptr) {
    const size_t G = gridDim.x;
    for (int j = threadIdx.x; j < d; j += PBLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NW; ++s2) s += gacc[(size_t)s2 * d + j];
      g_part[(size_t)j * G + blockIdx.x] = s;
    }
  } else {
    for (int j = threadIdx.x; j < d; j += PBLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NW; ++s2) s += gacc[(size_t)s2 * d + j];
      if (s != 0.f) atomicAdd(&g_out[j], s);
    }
  }
  if (lane == 0 && local_count) atomicAdd(n_out, local_count);
}

// Sums the per-block partial slabs into g (layout g_part[j][G], contiguous
// per column). Grid = ceil(d/BLOCK) * SPLITS.
__global__ __launch_bounds__(BLOCK) void reduce_partials_kernel(
    const float* __restrict__ g_part, float* __restrict__ g_out, int d,
    int G, int splits) {
  const int njc = (d + BLOCK - 1) / BLOCK;
  const int jc = blockIdx.x % njc;
  const int sp = blockIdx.x / njc;
  const int j = jc * BLOCK + threadIdx.x;
  if (j >= d) return;
  const int per = (G + splits - 1) / splits;
  const int b0 = sp * per;
  const int b1 = min(G, b0 + per);
  const float* base = g_part + (size_t)j * G;
  float s = 0.f;
  if (((b1 - b0) & 3) == 0 && (b0 & 3) == 0) {
    const float4* v = reinterpret_cast<const float4*>(base + b0);
    const int n4 = (b1 - b0) >> 2;
    for (int q = 0; q < n4; ++q) {
      const float4 x = v[q];
      s += x.x + x.y + x.z + x.w;
    }
  } else {
    for (int b = b0; b < b1; ++b) s += base[b];
  }
  if (splits == 1) g_out[j] += s;
  else atomicAdd(&g_out[j], s);
}

// ---------------------------------------------------------------- K2 (+K3)

template <typename VT, bool SAGA>
__global__ __launch_bounds__(BLOCK) void grad_csr_kernel(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const VT* __restrict__ values, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    int* __restrict__ n_out, float* __restrict__ alpha,
    int* __restrict__ idx_out, float* __restrict__ e_out,
    int* __restrict__ pos_ctr, const int* __restrict__ k_dev,
    int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, uint32_t threshold, int take_all, int objective) {
  if (k_dev) round_k = (uint32_t)(*k_dev) + 1u;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  int local_count = 0;
  const long gstride = (long)gridDim.x * ROWS_PER_BLOCK_ITER;
  for (long bb = (long)blockIdx.x * ROWS_PER_BLOCK_ITER; bb < n_rows;
       bb += gstride) {
    const long base = bb + (long)wave * ROWS_PER_WAVE;
    if (base >= n_rows) continue;
    const uint4 x = philox_block4(
        seed, round_k, (row_start + (uint64_t)base) / 4 + (uint64_t)lane);
    const long rem = n_rows - base;
    unsigned long long m[4];
    {
      const long lrow = 4L * lane;
      m[0] = __ballot(lrow + 0 < rem && (take_all || x.x < threshold));
      m[1] = __ballot(lrow + 1 < rem && (take_all || x.y < threshold));
      m[2] = __ballot(lrow + 2 < rem && (take_all || x.z < threshold));
      m[3] = __ballot(lrow + 3 < rem && (take_all || x.w < threshold));
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      unsigned long long mm = m[i];
      while (mm) {
        const int bit = __ffsll((long long)mm) - 1;
        mm &= mm - 1;
        const long rr = base + 4L * bit + i;
        const int s = indptr[rr], t = indptr[rr + 1];
        float z = 0.f;
        for (int p = s + lane; p < t; p += WAVE)
          z += to_f32<VT>(values[p]) * w[indices[p]];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) z += __shfl_xor(z, off, WAVE);
        float e = link_residual(z, y[rr], objective);
        float coeff = e;
        if (SAGA) {
          const float a_old = alpha[rr];
          coeff = e - a_old;
          if (lane == 0) {
            if (commit_now) {
              alpha[rr] = e;
            } else {
              const int pos = atomicAdd(pos_ctr, 1);
              idx_out[pos] = (int)rr;
              e_out[pos] = e;
            }
          }
        }
        ++local_count;
        for (int p = s + lane; p < t; p += WAVE)
          atomicAdd(&g_out[indices[p]], coeff * to_f32<VT>(values[p]));
      }
    }
  }
  if (lane == 0 && local_count) atomicAdd(n_out, local_count);
}

// ---------------------------------------------------------------- K5/K6

__global__ void sgd_update_kernel(float* __restrict__ w,
                                  const float* __restrict__ g, float gamma_k,
                                  float inv_batch, int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) w[i] -= gamma_k * inv_batch * g[i];
}

// w -= gamma*(inv_batch*g + alpha_bar_old); alpha_bar += inv_N*g
// (order matches reference SparkASAGASync.scala:300-304 /
//  SparkASAGAThread.scala:217-220: w reads the OLD alpha_bar).
__global__ void saga_update_kernel(float* __restrict__ w,
                                   const float* __restrict__ g,
                                   float* __restrict__ alpha_bar, float gamma,
                                   float inv_batch, float inv_N, int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) {
    const float gi = g[i];
    w[i] -= gamma * (inv_batch * gi + alpha_bar[i]);
    alpha_bar[i] += inv_N * gi;
  }
}

__global__ void saga_commit_kernel(float* __restrict__ alpha,
                                   const int* __restrict__ idx,
                                   const float* __restrict__ e, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) alpha[idx[i]] = e[i];
}

// Fused device-loop update kernels (graph mode): ONE workgroup applies the
// update, zeroes the gradient accumulator for the next round, and advances
// the device round counter — so an unrolled sequence of
// [grad, reduce, update] node triples forms a complete hipGraph with no
// host logic.
__global__ __launch_bounds__(1024) void sgd_update_fused_kernel(
    float* __restrict__ w, float* __restrict__ g, int* __restrict__ k_dev,
    float gamma, float inv_batch, int num_part, int d) {
  const int k = *k_dev;
  // integer division k/num_part matches the reference's Scala Int semantics
  // (SparkASGDThread.scala:190)
  const float gamma_k =
      (float)((double)gamma / sqrt((double)(k / num_part + 1)));
  for (int i = threadIdx.x; i < d; i += blockDim.x) {
    w[i] -= gamma_k * inv_batch * g[i];
    g[i] = 0.f;
  }
  __syncthreads();
  if (threadIdx.x == 0) *k_dev = k + 1;
}

__global__ __launch_bounds__(1024) void saga_update_fused_kernel(
    float* __restrict__ w, float* __restrict__ g,
    float* __restrict__ alpha_bar, int* __restrict__ k_dev, float gamma,
    float inv_batch, float inv_N, int d) {
  for (int i = threadIdx.x; i < d; i += blockDim.x) {
    const float gi = g[i];
    w[i] -= gamma * (inv_batch * gi + alpha_bar[i]);
    alpha_bar[i] += inv_N * gi;
    g[i] = 0.f;
  }
  __syncthreads();
  if (threadIdx.x == 0) *k_dev += 1;
}

// ---------------------------------------------------------------- launchers

static inline int grad_grid(long n_rows) {
  const char* s = std::getenv("ASYNCAMD_GRAD_GRID");  // re-read: sweeps
  const int override_grid = s ? std::atoi(s) : 0;
  if (override_grid > 0) return override_grid;
  long g = (n_rows + ROWS_PER_BLOCK_ITER - 1) / ROWS_PER_BLOCK_ITER;
  if (g > 512) g = 512;  // measured optimum on 8.1M rows (pipe kernel)
  if (g < 1) g = 1;
  return (int)g;
}

static inline int pick_lpr(int d) {
  const char* s = std::getenv("ASYNCAMD_LPR");
  const int override_lpr = s ? std::atoi(s) : 0;
  if (override_lpr == 64 || override_lpr == 32 || override_lpr == 16)
    return override_lpr;
  // measured on the mnist8m shape: LPR 64 > 32 > 16 (sub-wave rows cost LDS
  // occupancy more than the extra MLP buys) — keep whole-wave rows
  return 64;
}

template <typename XT, bool SAGA>
static void launch_dense(const XT* X, const float* y, const float* w,
                         float* g_out, float* g_part, int* n_out,
                         float* alpha, int* idx_out, float* e_out,
                         int* pos_ctr, const int* k_dev, int commit_now,
                         long n_rows, int d, uint64_t seed, uint32_t round_k,
                         uint64_t row_start, double rate, int objective,
                         hipStream_t stream) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  const char* np = std::getenv("ASYNCAMD_NO_PIPE");
  const bool pipe_ok = (d % 4 == 0) && (d <= 4 * WAVE * PIPE_MAXIT) &&
                       !(np && np[0] == '1');
  if (pipe_ok) {
    const char* pb = std::getenv("ASYNCAMD_PBLOCK");
    const int pblock = pb ? std::atoi(pb) : 256;  // measured best
#define LAUNCH_PIPE(PB, DP)                                                  \
    do {                                                                     \
      const size_t smem = (size_t)(1 + PB / WAVE) * d * sizeof(float) +      \
                          (QCAP_FOR(PB) + 1) * sizeof(int);                  \
      hipLaunchKernelGGL((grad_dense_pipe_kernel<XT, SAGA, PB, DP>),         \
                         dim3(grid), dim3(PB), smem, stream, X, y, w, g_out, \
                         g_part, n_out, alpha, idx_out, e_out, pos_ctr,      \
                         k_dev, commit_now, n_rows, d, seed, round_k,        \
                         row_start, thr, take_all, objective);               \
    } while (0)
    if (pblock == 256) LAUNCH_PIPE(256, 4);
    else LAUNCH_PIPE(512, 4);
#undef LAUNCH_PIPE
    return;
  }
  const int lpr = pick_lpr(d);
  const size_t smem = (size_t)(1 + 4 * (WAVE / lpr)) * d * sizeof(float);
#define DISPATCH_LPR(L)                                                      \
  hipLaunchKernelGGL((grad_dense_kernel<XT, SAGA, L>), dim3(grid),           \
                     dim3(BLOCK), smem, stream, X, y, w, g_out, g_part,      \
                     n_out, alpha, idx_out, e_out, pos_ctr, k_dev,           \
                     commit_now, n_rows, d, seed, round_k, row_start, thr,   \
                     take_all, objective)
  if (lpr == 16) DISPATCH_LPR(16);
  else if (lpr == 32) DISPATCH_LPR(32);
  else DISPATCH_LPR(64);
#undef DISPATCH_LPR
}

extern "C" {

int query_grad_grid(long n_rows) { return grad_grid(n_rows); }

void launch_grad_dense(const void* X, const float* y, const float* w,
                       float* g_out, float* g_part, int* n_out,
                       const int* k_dev, long n_rows, int d, uint64_t seed,
                       uint32_t round_k, uint64_t row_start, double rate,
                       int objective, int x_is_bf16, hipStream_t stream) {
  if (x_is_bf16)
    launch_dense<__hip_bfloat16, false>(
        (const __hip_bfloat16*)X, y, w, g_out, g_part, n_out, nullptr,
        nullptr, nullptr, nullptr, k_dev, 0, n_rows, d, seed, round_k,
        row_start, rate, objective, stream);
  else
    launch_dense<float, false>((const float*)X, y, w, g_out, g_part, n_out,
                               nullptr, nullptr, nullptr, nullptr, k_dev, 0,
                               n_rows, d, seed, round_k, row_start, rate,
                               objective, stream);
}

void launch_saga_grad_dense(const void* X, const float* y, const float* w,
                            float* alpha, float* g_out, float* g_part,
                            int* n_out, int* idx_out, float* e_out,
                            int* pos_ctr, const int* k_dev, int commit_now,
                            long n_rows, int d, uint64_t seed,
                            uint32_t round_k, uint64_t row_start, double rate,
                            int objective, int x_is_bf16,
                            hipStream_t stream) {
  if (x_is_bf16)
    launch_dense<__hip_bfloat16, true>(
        (const __hip_bfloat16*)X, y, w, g_out, g_part, n_out, alpha, idx_out,
        e_out, pos_ctr, k_dev, commit_now, n_rows, d, seed, round_k,
        row_start, rate, objective, stream);
  else
    launch_dense<float, true>((const float*)X, y, w, g_out, g_part, n_out,
                              alpha, idx_out, e_out, pos_ctr, k_dev,
                              commit_now, n_rows, d, seed, round_k, row_start,
                              rate, objective, stream);
}

void launch_reduce_partials(const float* g_part, float* g_out, int d, int G,
                            int splits, hipStream_t stream) {
  const int njc = (d + BLOCK - 1) / BLOCK;
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(njc * splits), dim3(BLOCK),
                     0, stream, g_part, g_out, d, G, splits);
}

void launch_grad_csr(const int* indptr, const int* indices, const void* values,
                     const float* y, const float* w, float* g_out, int* n_out,
                     const int* k_dev, long n_rows, uint64_t seed,
                     uint32_t round_k, uint64_t row_start, double rate,
                     int objective, int v_is_bf16, hipStream_t stream) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  if (v_is_bf16) {
    hipLaunchKernelGGL((grad_csr_kernel<__hip_bfloat16, false>), dim3(grid),
                       dim3(BLOCK), 0, stream, indptr, indices,
                       (const __hip_bfloat16*)values, y, w, g_out, n_out,
                       nullptr, nullptr, nullptr, nullptr, k_dev, 0, n_rows,
                       seed, round_k, row_start, thr, take_all, objective);
  } else {
    hipLaunchKernelGGL((grad_csr_kernel<float, false>), dim3(grid),
                       dim3(BLOCK), 0, stream, indptr, indices,
                       (const float*)values, y, w, g_out, n_out, nullptr,
                       nullptr, nullptr, nullptr, k_dev, 0, n_rows, seed,
                       round_k, row_start, thr, take_all, objective);
  }
}

void launch_saga_grad_csr(const int* indptr, const int* indices,
                          const void* values, const float* y, const float* w,
                          float* alpha, float* g_out, int* n_out,
                          int* idx_out, float* e_out, int* pos_ctr,
                          const int* k_dev, int commit_now, long n_rows,
                          uint64_t seed, uint32_t round_k, uint64_t row_start,
                          double rate, int objective, int v_is_bf16,
                          hipStream_t stream) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  if (v_is_bf16) {
    hipLaunchKernelGGL((grad_csr_kernel<__hip_bfloat16, true>), dim3(grid),
                       dim3(BLOCK), 0, stream, indptr, indices,
                       (const __hip_bfloat16*)values, y, w, g_out, n_out,
                       alpha, idx_out, e_out, pos_ctr, k_dev, commit_now,
                       n_rows, seed, round_k, row_start, thr, take_all,
                       objective);
  } else {
    hipLaunchKernelGGL((grad_csr_kernel<float, true>), dim3(grid),
                       dim3(BLOCK), 0, stream, indptr, indices,
                       (const float*)values, y, w, g_out, n_out, alpha,
                       idx_out, e_out, pos_ctr, k_dev, commit_now, n_rows,
                       seed, round_k, row_start, thr, take_all, objective);
  }
}

void launch_sgd_update(float* w, const float* g, float gamma_k,
                       float inv_batch, int d, hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(sgd_update_kernel, dim3(grid), dim3(256), 0, stream, w,
                     g, gamma_k, inv_batch, d);
}

void launch_saga_update(float* w, const float* g, float* alpha_bar,
                        float gamma, float inv_batch, float inv_N, int d,
                        hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(saga_update_kernel, dim3(grid), dim3(256), 0, stream, w,
                     g, alpha_bar, gamma, inv_batch, inv_N, d);
}

void launch_saga_commit(float* alpha, const int* idx, const float* e, int n,
                        hipStream_t stream) {
  if (n <= 0) return;
  const int grid = (n + 255) / 256;
  hipLaunchKernelGGL(saga_commit_kernel, dim3(grid), dim3(256), 0, stream,
                     alpha, idx, e, n);
}

void launch_sgd_update_fused(float* w, float* g, int* k_dev, float gamma,
                             float inv_batch, int num_part, int d,
                             hipStream_t stream) {
  hipLaunchKernelGGL(sgd_update_fused_kernel, dim3(1), dim3(1024), 0, stream,
                     w, g, k_dev, gamma, inv_batch, num_part, d);
}

void launch_saga_update_fused(float* w, float* g, float* alpha_bar,
                              int* k_dev, float gamma, float inv_batch,
                              float inv_N, int d, hipStream_t stream) {
  hipLaunchKernelGGL(saga_update_fused_kernel, dim3(1), dim3(1024), 0,
                     stream, w, g, alpha_bar, k_dev, gamma, inv_batch, inv_N,
                     d);
}

}  // extern "C"
