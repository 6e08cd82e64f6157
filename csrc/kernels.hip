// Hand-written CDNA4 (gfx950 / MI355X) kernels for the ASYNC hot path.
//
// Kernel inventory (reference JVM hot loops they replace):
//   K1 grad_dense      — fused Philox sample mask + per-row dot + scaled
//                        accumulate (reference gradfun SparkASGDThread.scala:
//                        423-438 + reducePartition fold RDD.scala:1103-1123);
//                        three forms: grad_dense_pipe_kernel (LDS row queue
//                        + depth-4 register pipeline), scan_rows_kernel +
//                        grad_dense_list_kernel (split form for the graph
//                        engine's scan/compute overlap), and a generic
//                        fallback for d > 2048 or d % 4 != 0.
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
#include "grad_wave.h"

#define WAVE 64
#define BLOCK 256
#define WAVES_PER_BLOCK (BLOCK / WAVE)
#define ROWS_PER_WAVE 256  // 64 lanes x 4 rows per philox eval
#define ROWS_PER_BLOCK_ITER (WAVES_PER_BLOCK * ROWS_PER_WAVE)

// ---------------------------------------------------------------- helpers

// Host-visible completion flag: the LAST arriving block publishes
// `done_val` to fine-grained pinned host memory with system-scope release,
// replacing the native engine's hipEventRecord + hipEventQuery pair
// (~2-4 us of host API per round). All prior global writes (g atomics,
// SAGA staging) are ordered before the flag by the fence chain.
__device__ __forceinline__ void publish_done(
    unsigned long long* done_flag, unsigned long long done_val,
    unsigned long long* done_arr, unsigned int nblk) {
  if (!done_flag) return;  // uniform per launch: no divergence
  // NO fences here — fence instructions at agent scope on a multi-XCD
  // chip emit per-block L2 writeback/invalidate (buffer_wbl2/inv), which
  // poisoned every concurrent kernel's L2 (measured: grad 16.6 -> 117 us
  // with a __threadfence + acq_rel arrival). The gradient's cross-kernel
  // outputs (g, n_out) are written with ATOMICS, which execute at the
  // agent coherence point; all that is needed before arrival is to DRAIN
  // this thread's outstanding vmem ops, then count with a relaxed RMW.
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned long long arrived = __hip_atomic_fetch_add(
        done_arr, 1ull, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (arrived + 1 == (unsigned long long)nblk) {
      // reset for the next round with a RETURNING exchange, then drain it
      // before the flag store: guarantees the reset is globally ordered
      // before the host can observe this round's completion
      (void)__hip_atomic_exchange(done_arr, 0ull, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
      __builtin_amdgcn_s_waitcnt(0);
      __hip_atomic_store(done_flag, done_val, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
}

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
    int take_all, int objective, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr) {
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
              // atomic staging: read cross-stream in wave mode
              const int pos = atomicAdd(pos_ctr, 1);
              __hip_atomic_exchange(&idx_out[pos], (int)rr,
                                    __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
              __hip_atomic_exchange(&e_out[pos], e, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
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
  publish_done(done_flag, done_val, done_arr, gridDim.x);
}

// ------------------------------------------------- K1 pipelined (queue)
//
// Two-phase dense gradient. Phase 1 (scan): Philox decides 4 rows per eval;
// sampled rows are appended to an LDS queue TOGETHER with their y (and SAGA
// alpha) values — prefetching them here keeps ordinary vmem loads out of
// the pipelined phase (hipcc drains vmcnt(0) at any ordinary-load use,
// CDNA guide §5.5 trap (b): measured one full drain per row from y[rr]).
// Phase 2 (process): waves drain the queue round-robin with a DEPTH-deep
// statically-unrolled register pipeline; row loads are UNCONDITIONAL
// bounds-checked buffer loads (per-row scalar descriptor) with a
// compile-time ITERS count — a per-element `if (j4 < d4)` made hipcc branch
// around every load with a vmcnt(0) (guide trap 4(c)). The accumulate pass
// reuses the pipeline registers; X is read exactly once per sampled row.

#define QCAP 2048

// Row-fragment types for the pipelined loaders. Loads are compiler-visible
// buffer intrinsics: with unconditional compile-time load counts and
// sched_barrier(0) fences around the refills (see the pipe kernels), hipcc
// emits a counted descending vmcnt ladder — a genuine software pipeline
// with compiler-guaranteed hazards (hand-counted inline-asm waits were
// tried and abandoned: regalloc preservation copies raced in-flight asm
// loads).
struct B16x4 { union { ushort4 s; uint2 u; }; };
struct F32x4 { union { float4 f; uint4 u; }; };

template <typename XT> struct RowVec;
template <> struct RowVec<float> {
  using T = F32x4;
  static __device__ __forceinline__ T load(__amdgpu_buffer_rsrc_t rsrc,
                                           int voff_bytes) {
    T out;
    auto r = __builtin_amdgcn_raw_buffer_load_b128(rsrc, voff_bytes, 0, 0);
    union { decltype(r) u; float4 f; } c{r};
    out.f = c.f;
    return out;
  }
  static constexpr int VOFF_SHIFT = 4;  // 16 B per lane-element
};
template <> struct RowVec<__hip_bfloat16> {
  using T = B16x4;
  static __device__ __forceinline__ T load(__amdgpu_buffer_rsrc_t rsrc,
                                           int voff_bytes) {
    T out;
    auto r = __builtin_amdgcn_raw_buffer_load_b64(rsrc, voff_bytes, 0, 0);
    union { decltype(r) u; ushort4 s; } c{r};
    out.s = c.s;
    return out;
  }
  static constexpr int VOFF_SHIFT = 3;  // 8 B per lane-element
};

__device__ __forceinline__ void cvt4(const float4& r, float o[4]) {
  o[0] = r.x; o[1] = r.y; o[2] = r.z; o[3] = r.w;
}
__device__ __forceinline__ void cvt4(const ushort4& r, float o[4]) {
  union { unsigned short u; __hip_bfloat16 b; } c0{r.x}, c1{r.y}, c2{r.z},
      c3{r.w};
  o[0] = __bfloat162float(c0.b); o[1] = __bfloat162float(c1.b);
  o[2] = __bfloat162float(c2.b); o[3] = __bfloat162float(c3.b);
}
__device__ __forceinline__ void cvt4(const F32x4& r, float o[4]) {
  cvt4(r.f, o);
}
__device__ __forceinline__ void cvt4(const B16x4& r, float o[4]) {
  cvt4(r.s, o);
}

// Per-row buffer descriptor: base = row start (scalar via readfirstlane),
// num_records = bytes to the end of X -> hardware bounds check returns 0 for
// the padded over-read of the last row (w_lds is zero-padded, so padded
// elements contribute nothing).
template <typename XT>
__device__ __forceinline__ __amdgpu_buffer_rsrc_t row_rsrc(
    const XT* X, long total_elems, int rr, int d) {
  const size_t off = (size_t)rr * d;
  const uint64_t rem_bytes = (uint64_t)(total_elems - off) * sizeof(XT);
  const uint32_t nrec =
      rem_bytes > 0xFFFFFFF0ull ? 0xFFFFFFF0u : (uint32_t)rem_bytes;
  return __builtin_amdgcn_make_buffer_rsrc((void*)(X + off), 0, nrec,
                                           0x00020000);
}

// Standalone mask scan: fills a GLOBAL compacted row list (+ prefetched y
// values) for a FUTURE round. The scan depends only on (seed, round), not on
// w — so the graph engine runs round r+1's scan on a side stream overlapped
// with round r's gradient/update. The round key comes from a dedicated
// scan_round counter bumped stream-order by bump_counter_kernel (reading
// k_dev here would race the concurrent update's k++).
__device__ void scan_rows_body(
    const float* __restrict__ y, int* __restrict__ rowlist,
    float* __restrict__ ylist, int* __restrict__ count_dev,
    const int* __restrict__ scan_round_dev, long n_rows, uint64_t seed,
    uint32_t round_k, uint64_t row_start, uint32_t threshold, int take_all,
    int bid, int nblk) {
  if (scan_round_dev) round_k = (uint32_t)(*scan_round_dev);
  constexpr int SCAP = 2048;
  __shared__ int rows_s[SCAP];
  __shared__ int qn_s, base_s;
  if (threadIdx.x == 0) qn_s = 0;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const long ngroups = (n_rows + ROWS_PER_BLOCK_ITER - 1) / ROWS_PER_BLOCK_ITER;
  long gi = bid;
  bool done = false;
  while (!done) {
    while (true) {
      __syncthreads();
      if (gi >= ngroups || qn_s >= SCAP - ROWS_PER_BLOCK_ITER) break;
      const long base = gi * (long)ROWS_PER_BLOCK_ITER +
                        (long)wave * ROWS_PER_WAVE;
      if (base < n_rows) {
        const uint4 x = philox_block4(
            seed, round_k, (row_start + (uint64_t)base) / 4 + (uint64_t)lane);
        const long rem = n_rows - base;
        const long lrow = base + 4L * lane;
        const uint32_t xs[4] = {x.x, x.y, x.z, x.w};
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (4L * lane + i < rem && (take_all || xs[i] < threshold)) {
            const int pos = atomicAdd(&qn_s, 1);
            rows_s[pos] = (int)(lrow + i);
          }
        }
      }
      gi += nblk;
    }
    __syncthreads();
    const int nq = min(qn_s, SCAP);
    if (threadIdx.x == 0 && nq > 0) base_s = atomicAdd(count_dev, nq);
    __syncthreads();
    for (int i = threadIdx.x; i < nq; i += BLOCK) {
      const int r = rows_s[i];
      rowlist[base_s + i] = r;
      ylist[base_s + i] = y[r];
    }
    __syncthreads();
    if (threadIdx.x == 0) qn_s = 0;
    done = gi >= ngroups;
  }
}

__global__ __launch_bounds__(BLOCK) void scan_rows_kernel(
    const float* __restrict__ y, int* __restrict__ rowlist,
    float* __restrict__ ylist, int* __restrict__ count_dev,
    const int* __restrict__ scan_round_dev, long n_rows, uint64_t seed,
    uint32_t round_k, uint64_t row_start, uint32_t threshold, int take_all) {
  scan_rows_body(y, rowlist, ylist, count_dev, scan_round_dev, n_rows, seed,
                 round_k, row_start, threshold, take_all, blockIdx.x,
                 gridDim.x);
}

__global__ void bump_counter_kernel(int* __restrict__ p) { *p += 1; }

// List-fed gradient: the process phase of the pipe kernel, consuming a
// globally compacted (rowlist, ylist) produced by scan_rows_kernel. Each
// block stages its contiguous slice into LDS first (per-row ordinary
// global loads inside the pipeline would reintroduce the vmcnt(0) drains).
template <typename XT, int PBLOCK, int DEPTH, int ITERS>
__global__ __launch_bounds__(PBLOCK) void grad_dense_list_kernel(
    const XT* __restrict__ X, const float* __restrict__ w,
    float* __restrict__ g_part, const int* __restrict__ rowlist,
    const float* __restrict__ ylist, const int* __restrict__ count_dev,
    long n_rows, int d, int objective) {
  constexpr int NW = PBLOCK / WAVE;
  constexpr int DPAD = ITERS * 256;
  constexpr int LCAP = 2048;
  extern __shared__ float smem[];
  float* w_lds = smem;                                // [DPAD]
  float* gacc = smem + DPAD;                          // [NW][DPAD]
  float* yq = smem + (size_t)(1 + NW) * DPAD;         // [LCAP]
  int* rowq = (int*)(yq + LCAP);                      // [LCAP]
  for (int j = threadIdx.x; j < DPAD; j += PBLOCK) {
    w_lds[j] = j < d ? w[j] : 0.f;
#pragma unroll
    for (int s2 = 0; s2 < NW; ++s2) gacc[(size_t)s2 * DPAD + j] = 0.f;
  }
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float* gw = gacc + (size_t)wave * DPAD;
  const long total_elems = n_rows * (long)d;
  using RV = RowVec<XT>;
  const int total = *count_dev;
  const int per_block = (total + gridDim.x - 1) / gridDim.x;
  const int my0 = blockIdx.x * per_block;
  const int my1 = min(total, my0 + per_block);
  for (int off = my0; off < my1; off += LCAP) {
    const int nq = min(my1 - off, LCAP);
    __syncthreads();
    for (int i = threadIdx.x; i < nq; i += PBLOCK) {
      rowq[i] = rowlist[off + i];
      yq[i] = ylist[off + i];
    }
    __syncthreads();
    typename RV::T buf[DEPTH][ITERS];
    if (nq > 0) {
#pragma unroll
      for (int p = 0; p < DEPTH; ++p) {
        const int q = min(wave + p * NW, nq - 1);
        const int rr = __builtin_amdgcn_readfirstlane(rowq[q]);
        const auto rs = row_rsrc<XT>(X, total_elems, rr, d);
#pragma unroll
        for (int it = 0; it < ITERS; ++it)
          buf[p][it] = RV::load(rs, (lane + it * WAVE) << RV::VOFF_SHIFT);
      }
    }
    int q_base = wave;
    while (q_base < nq) {
#pragma unroll
      for (int p = 0; p < DEPTH; ++p) {
        const int q = q_base + p * NW;
        if (q < nq) {
          float z = 0.f;
#pragma unroll
          for (int it = 0; it < ITERS; ++it) {
            float o[4];
            cvt4(buf[p][it], o);
            const int j4 = lane + it * WAVE;
            const float4 wv = reinterpret_cast<const float4*>(w_lds)[j4];
            z += o[0] * wv.x + o[1] * wv.y + o[2] * wv.z + o[3] * wv.w;
          }
#pragma unroll
          for (int offx = 32; offx > 0; offx >>= 1)
            z += __shfl_xor(z, offx, WAVE);
          const float coeff = link_residual(z, yq[q], objective);
#pragma unroll
          for (int it = 0; it < ITERS; ++it) {
            float o[4];
            cvt4(buf[p][it], o);
            const int j4 = lane + it * WAVE;
            float4* gw4 = reinterpret_cast<float4*>(gw);
            float4 cur = gw4[j4];
            cur.x += coeff * o[0]; cur.y += coeff * o[1];
            cur.z += coeff * o[2]; cur.w += coeff * o[3];
            gw4[j4] = cur;
          }
          __builtin_amdgcn_sched_barrier(0);
          const int qf = min(q + DEPTH * NW, nq - 1);
          {
            const int rr2 = __builtin_amdgcn_readfirstlane(rowq[qf]);
            const auto rs2 = row_rsrc<XT>(X, total_elems, rr2, d);
#pragma unroll
            for (int it = 0; it < ITERS; ++it)
              buf[p][it] = RV::load(rs2, (lane + it * WAVE) << RV::VOFF_SHIFT);
          }
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      q_base += DEPTH * NW;
    }
  }
  __syncthreads();
  const size_t G = gridDim.x;
  for (int j = threadIdx.x; j < d; j += PBLOCK) {
    float s = 0.f;
#pragma unroll
    for (int s2 = 0; s2 < NW; ++s2) s += gacc[(size_t)s2 * DPAD + j];
    g_part[(size_t)j * G + blockIdx.x] = s;
  }
}

template <typename XT, bool SAGA, int PBLOCK, int DEPTH, int ITERS>
__device__ void grad_dense_pipe_body(
    const XT* __restrict__ X, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    float* __restrict__ g_part, int* __restrict__ n_out,
    float* __restrict__ alpha, int* __restrict__ idx_out,
    float* __restrict__ e_out, int* __restrict__ pos_ctr,
    const int* __restrict__ k_dev, int commit_now, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, uint32_t threshold,
    int take_all, int objective, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr, int bid,
    int nblk) {
  if (k_dev) round_k = (uint32_t)(*k_dev) + 1u;
  constexpr int NW = PBLOCK / WAVE;
  constexpr int RPB = NW * ROWS_PER_WAVE;
  constexpr int DPAD = ITERS * 256;  // padded feature dim
  static_assert(QCAP >= RPB + 1024, "queue must out-size one scan iter");
  extern __shared__ float smem[];
  float* w_lds = smem;                               // [DPAD]
  float* gacc = smem + DPAD;                         // [NW][DPAD]
  float* yq = smem + (size_t)(1 + NW) * DPAD;        // [QCAP]
  float* aq = yq + QCAP;                             // [QCAP] (SAGA)
  float* eq = aq + (SAGA ? QCAP : 0);                // [QCAP] (SAGA)
  int* rowq = (int*)(eq + (SAGA ? QCAP : 0));        // [QCAP]
  int* qn = rowq + QCAP;
  for (int j = threadIdx.x; j < DPAD; j += PBLOCK) {
    w_lds[j] = j < d ? w[j] : 0.f;
#pragma unroll
    for (int s2 = 0; s2 < NW; ++s2) gacc[(size_t)s2 * DPAD + j] = 0.f;
  }
  if (threadIdx.x == 0) *qn = 0;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float* gw = gacc + (size_t)wave * DPAD;
  int local_count = 0;
  const long total_elems = n_rows * (long)d;
  using RV = RowVec<XT>;

  const long ngroups = (n_rows + RPB - 1) / RPB;
  long gi = bid;
  bool done = false;
  while (!done) {
    // ---- scan phase (also prefetches y / alpha into the queue)
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
            const int row = (int)(lrow + i);
            rowq[pos] = row;
            yq[pos] = y[row];
            if (SAGA) aq[pos] = alpha[row];
          }
        }
      }
      gi += nblk;
    }
    __syncthreads();
    const int nq = min(*qn, QCAP);
    // ---- process phase: DEPTH-deep static pipeline, buffer loads only
    typename RV::T buf[DEPTH][ITERS];
    // the counted-wait protocol needs EXACTLY ITERS loads per slot: a
    // skipped load would let vmcnt(N) pass while this buffer's own loads
    // are still in flight (vmcnt waits "<= N outstanding", nothing more).
    // Tail slots load a clamped row instead of skipping.
    if (nq > 0) {
#pragma unroll
      for (int p = 0; p < DEPTH; ++p) {
        const int q = min(wave + p * NW, nq - 1);
        const int rr = __builtin_amdgcn_readfirstlane(rowq[q]);
        const auto rs = row_rsrc<XT>(X, total_elems, rr, d);
#pragma unroll
        for (int it = 0; it < ITERS; ++it)
          buf[p][it] = RV::load(rs, (lane + it * WAVE) << RV::VOFF_SHIFT);
      }
    }
    int q_base = wave;
    while (q_base < nq) {
#pragma unroll
      for (int p = 0; p < DEPTH; ++p) {
        const int q = q_base + p * NW;
        if (q < nq) {
          float z = 0.f;
#pragma unroll
          for (int it = 0; it < ITERS; ++it) {
            float o[4];
            cvt4(buf[p][it], o);
            const int j4 = lane + it * WAVE;
            const float4 wv = reinterpret_cast<const float4*>(w_lds)[j4];
            z += o[0] * wv.x + o[1] * wv.y + o[2] * wv.z + o[3] * wv.w;
          }
#pragma unroll
          for (int off = 32; off > 0; off >>= 1)
            z += __shfl_xor(z, off, WAVE);
          float e = link_residual(z, yq[q], objective);
          float coeff = e;
          if (SAGA) {
            coeff = e - aq[q];
            if (lane == 0) eq[q] = e;  // committed after the barrier
          }
          ++local_count;
#pragma unroll
          for (int it = 0; it < ITERS; ++it) {
            float o[4];
            cvt4(buf[p][it], o);
            const int j4 = lane + it * WAVE;
            float4* gw4 = reinterpret_cast<float4*>(gw);
            float4 cur = gw4[j4];
            cur.x += coeff * o[0]; cur.y += coeff * o[1];
            cur.z += coeff * o[2]; cur.w += coeff * o[3];
            gw4[j4] = cur;
          }
          // refill this buffer DEPTH rows ahead (clamped: see prologue).
          // sched_barrier pins every read of buf[p] (the accumulate above)
          // BEFORE the async refill overwrites it — the compiler treats an
          // asm load's register write as instantaneous and would otherwise
          // rotate reads past the issue (observed: regalloc preservation
          // copies racing the in-flight load).
          __builtin_amdgcn_sched_barrier(0);
          const int qf = min(q + DEPTH * NW, nq - 1);
          {
            const int rr2 = __builtin_amdgcn_readfirstlane(rowq[qf]);
            const auto rs2 = row_rsrc<XT>(X, total_elems, rr2, d);
#pragma unroll
            for (int it = 0; it < ITERS; ++it)
              buf[p][it] = RV::load(rs2, (lane + it * WAVE) << RV::VOFF_SHIFT);
          }
          __builtin_amdgcn_sched_barrier(0);
        }
      }
      q_base += DEPTH * NW;
    }
    __syncthreads();
    if (SAGA) {
      // history scalar emit/commit, batched outside the pipeline
      for (int pos = threadIdx.x; pos < nq; pos += PBLOCK) {
        const int row = rowq[pos];
        const float e = eq[pos];
        if (commit_now) {
          alpha[row] = e;
        } else {
          // atomic staging: read cross-stream in wave mode
          const int gpos = atomicAdd(pos_ctr, 1);
          __hip_atomic_exchange(&idx_out[gpos], row, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
          __hip_atomic_exchange(&e_out[gpos], e, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
        }
      }
      __syncthreads();
    }
    if (threadIdx.x == 0) *qn = 0;
    done = gi >= ngroups;
  }
  __syncthreads();
  if (g_part != nullptr) {
    const size_t G = nblk;
    for (int j = threadIdx.x; j < d; j += PBLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NW; ++s2) s += gacc[(size_t)s2 * DPAD + j];
      g_part[(size_t)j * G + bid] = s;
    }
  } else {
    for (int j = threadIdx.x; j < d; j += PBLOCK) {
      float s = 0.f;
#pragma unroll
      for (int s2 = 0; s2 < NW; ++s2) s += gacc[(size_t)s2 * DPAD + j];
      if (s != 0.f) atomicAdd(&g_out[j], s);
    }
  }
  if (lane == 0 && local_count) atomicAdd(n_out, local_count);
  publish_done(done_flag, done_val, done_arr, nblk);
}

template <typename XT, bool SAGA, int PBLOCK, int DEPTH, int ITERS>
__global__ __launch_bounds__(PBLOCK) void grad_dense_pipe_kernel(
    const XT* __restrict__ X, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    float* __restrict__ g_part, int* __restrict__ n_out,
    float* __restrict__ alpha, int* __restrict__ idx_out,
    float* __restrict__ e_out, int* __restrict__ pos_ctr,
    const int* __restrict__ k_dev, int commit_now, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, uint32_t threshold,
    int take_all, int objective, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr) {
  grad_dense_pipe_body<XT, SAGA, PBLOCK, DEPTH, ITERS>(
      X, y, w, g_out, g_part, n_out, alpha, idx_out, e_out, pos_ctr, k_dev,
      commit_now, n_rows, d, seed, round_k, row_start, threshold, take_all,
      objective, done_flag, done_val, done_arr, blockIdx.x, gridDim.x);
}

// ---- wave launch: ONE kernel runs a whole quorum wave of dense-ASGD
// rounds (the native engine previously paid ~5 us of launch latency per
// worker per wave). Block b serves wave slot b / bper with intra-worker
// block id b % bper; each worker's sub-grid publishes its OWN done flag
// as it drains, so completions still stagger exactly as with per-worker
// kernels. Per-worker invariants (X, y, wbuf, g, n_rows, flags) live in a
// device table built once at engine init; the per-round variables (slot
// list, Philox round keys, serials) travel by value in the launch args.
template <typename XT, bool SAGA, int PBLOCK, int DEPTH, int ITERS>
__global__ __launch_bounds__(PBLOCK) void grad_dense_wave_kernel(
    const GradWaveSlot* __restrict__ slots, GradWaveCmd cmd, int d,
    uint64_t seed, uint32_t threshold, int take_all, int objective) {
  const int si = cmd.interleave ? (int)(blockIdx.x % cmd.n)
                                : (int)(blockIdx.x / cmd.bper);
  const int bid = cmd.interleave ? (int)(blockIdx.x / cmd.n)
                                 : (int)(blockIdx.x % cmd.bper);
  const GradWaveSlot sl = slots[cmd.wid[si]];
  grad_dense_pipe_body<XT, SAGA, PBLOCK, DEPTH, ITERS>(
      (const XT*)sl.X, sl.y, sl.wbuf, sl.g, nullptr, sl.n_out, sl.alpha,
      sl.idx_out, sl.e_out, sl.pos_ctr, nullptr, 0, sl.n_rows, d, seed,
      cmd.round_k[si], (uint64_t)sl.row_start, threshold, take_all,
      objective, sl.done_flag, cmd.done_val[si], sl.done_arr, bid,
      cmd.bper);
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

// LPR = lanes per row: rcv1-class rows have ~73 nnz, so a 64-lane row
// wastes most of the wave; 16-lane sub-waves keep 4 rows in flight per
// wave (the gradient scatter uses global atomics — no LDS slabs — so
// unlike the dense kernel, sub-waves here cost no occupancy).
template <typename VT, bool SAGA, int LPR>
__device__ void grad_csr_body(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const VT* __restrict__ values, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    int* __restrict__ n_out, float* __restrict__ alpha,
    int* __restrict__ idx_out, float* __restrict__ e_out,
    int* __restrict__ pos_ctr, const int* __restrict__ k_dev,
    int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, uint32_t threshold, int take_all, int objective,
    unsigned long long* done_flag, unsigned long long done_val,
    unsigned long long* done_arr, int bid, int nblk) {
  if (k_dev) round_k = (uint32_t)(*k_dev) + 1u;
  constexpr int NSUB = WAVE / LPR;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int sub = lane / LPR;
  const int sl = lane % LPR;
  int local_count = 0;
  const long gstride = (long)nblk * ROWS_PER_BLOCK_ITER;
  for (long bb = (long)bid * ROWS_PER_BLOCK_ITER; bb < n_rows;
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
      for (int s2 = 0; s2 < sub && mm; ++s2) mm &= mm - 1;  // my first bit
      while (mm) {
        const int bit = __ffsll((long long)mm) - 1;
        const long rr = base + 4L * bit + i;
        const int s = indptr[rr], t = indptr[rr + 1];
        float z = 0.f;
        for (int p = s + sl; p < t; p += LPR)
          z += to_f32<VT>(values[p]) * w[indices[p]];
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
              alpha[rr] = e;
            } else {
              // ATOMIC staging: the wave-mode commit kernel reads these
              // from another kernel with no stream-order edge, so the
              // writes must land at the agent coherence point (plain
              // stores can sit in one XCD's L2)
              const int pos = atomicAdd(pos_ctr, 1);
              __hip_atomic_exchange(&idx_out[pos], (int)rr,
                                    __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
              __hip_atomic_exchange(&e_out[pos], e, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
            }
          }
        }
        ++local_count;
        for (int p = s + sl; p < t; p += LPR)
          atomicAdd(&g_out[indices[p]], coeff * to_f32<VT>(values[p]));
        for (int s2 = 0; s2 < NSUB && mm; ++s2) mm &= mm - 1;  // next mine
      }
    }
  }
  if (sl == 0 && local_count) atomicAdd(n_out, local_count);
  publish_done(done_flag, done_val, done_arr, nblk);
}

template <typename VT, bool SAGA, int LPR>
__global__ __launch_bounds__(BLOCK) void grad_csr_kernel(
    const int* __restrict__ indptr, const int* __restrict__ indices,
    const VT* __restrict__ values, const float* __restrict__ y,
    const float* __restrict__ w, float* __restrict__ g_out,
    int* __restrict__ n_out, float* __restrict__ alpha,
    int* __restrict__ idx_out, float* __restrict__ e_out,
    int* __restrict__ pos_ctr, const int* __restrict__ k_dev,
    int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, uint32_t threshold, int take_all, int objective,
    unsigned long long* done_flag, unsigned long long done_val,
    unsigned long long* done_arr) {
  grad_csr_body<VT, SAGA, LPR>(indptr, indices, values, y, w, g_out, n_out,
                               alpha, idx_out, e_out, pos_ctr, k_dev,
                               commit_now, n_rows, seed, round_k, row_start,
                               threshold, take_all, objective, done_flag,
                               done_val, done_arr, blockIdx.x, gridDim.x);
}

// CSR wave: one kernel per quorum wave (see grad_dense_wave_kernel)
template <typename VT, bool SAGA, int LPR>
__global__ __launch_bounds__(BLOCK) void grad_csr_wave_kernel(
    const CsrWaveSlot* __restrict__ slots, GradWaveCmd cmd, uint64_t seed,
    uint32_t threshold, int take_all, int objective) {
  const int si = cmd.interleave ? (int)(blockIdx.x % cmd.n)
                                : (int)(blockIdx.x / cmd.bper);
  const int bid = cmd.interleave ? (int)(blockIdx.x / cmd.n)
                                 : (int)(blockIdx.x % cmd.bper);
  const CsrWaveSlot sl = slots[cmd.wid[si]];
  grad_csr_body<VT, SAGA, LPR>(
      sl.indptr, sl.indices, (const VT*)sl.values, sl.y, sl.wbuf, sl.g,
      sl.n_out, sl.alpha, sl.idx_out, sl.e_out, sl.pos_ctr, nullptr, 0,
      sl.n_rows, seed, cmd.round_k[si], (uint64_t)sl.row_start, threshold,
      take_all, objective, sl.done_flag, cmd.done_val[si], sl.done_arr, bid,
      cmd.bper);
}

// SAGA commit + staging-reset pass for a whole wave, launched before the
// wave's grad kernel on the same stream. Scatters the previous accepted
// round's staged scalars into the worker-resident alpha slice
// (commit-gated) and resets the staging counter unconditionally (a
// REJECTED previous round must also restart its staging list). Reads use
// RMW loads: the staging was written by another kernel with no
// stream-order edge, so normal loads could hit a stale L2 line.
__device__ __forceinline__ int atomic_load_int(int* p) {
  return __hip_atomic_fetch_add(p, 0, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ float atomic_load_f32(float* p) {
  return __hip_atomic_fetch_add(p, 0.f, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
}

__global__ __launch_bounds__(BLOCK) void saga_commit_wave_kernel(
    const CommitSlot* __restrict__ slots, CsrCommitCmd cmd) {
  const int si = (int)(blockIdx.x % cmd.n);
  const int bid = (int)(blockIdx.x / cmd.n);
  const CommitSlot sl = slots[cmd.wid[si]];
  const int cnt = atomic_load_int(sl.pos_ctr);
  if (cmd.do_commit[si]) {
    for (int i = bid * BLOCK + threadIdx.x; i < cnt;
         i += cmd.bper * BLOCK) {
      const int r = atomic_load_int(&sl.idx[i]);
      const float e = atomic_load_f32(&sl.e[i]);
      sl.dst[r] = e;  // only this worker's rounds touch its slice
    }
  }
  // last block of the slot resets the staging + sample counters (reuses
  // done_arr, which is 0 between rounds; the wave's grad kernel is
  // stream-ordered after this kernel)
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned long long a = __hip_atomic_fetch_add(
        sl.arr, 1ull, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (a + 1 == (unsigned long long)cmd.bper) {
      (void)__hip_atomic_exchange(sl.arr, 0ull, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
      (void)__hip_atomic_exchange(sl.pos_ctr, 0, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
      (void)__hip_atomic_exchange(sl.n_out, 0, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
      if (sl.scnt)
        (void)__hip_atomic_exchange(sl.scnt, 0, __ATOMIC_RELAXED,
                                    __HIP_MEMORY_SCOPE_AGENT);
      __builtin_amdgcn_s_waitcnt(0);
    }
  }
}

// spill-refresh wave: per slot, recompute the round's Philox row set into
// (srows, scnt), then gather ONLY those entries from the pinned master
// into the device staging table (both stream-ordered before the wave's
// gradient). Slot-mapped wrappers over the singleton kernels' logic.
__global__ __launch_bounds__(BLOCK) void scan_rows_wave_kernel(
    const GradWaveSlot* __restrict__ slots, GradWaveCmd cmd, uint64_t seed,
    uint32_t threshold, int take_all) {
  const int si = (int)(blockIdx.x % cmd.n);
  const int bid = (int)(blockIdx.x / cmd.n);
  const GradWaveSlot sl = slots[cmd.wid[si]];
  scan_rows_body(sl.y, sl.srows, sl.sylist, sl.scnt, nullptr, sl.n_rows,
                 seed, cmd.round_k[si], (uint64_t)sl.row_start, threshold,
                 take_all, bid, cmd.bper);
}

__global__ void alpha_gather_wave_kernel(
    const GradWaveSlot* __restrict__ slots, GradWaveCmd cmd) {
  const int si = (int)(blockIdx.x % cmd.n);
  const int bid = (int)(blockIdx.x / cmd.n);
  const GradWaveSlot sl = slots[cmd.wid[si]];
  const int n = *sl.scnt;  // same stream as the scan: plain read is fine
  for (int i = bid * BLOCK + threadIdx.x; i < n; i += cmd.bper * BLOCK) {
    const int r = sl.srows[i];
    sl.alpha[r] = sl.alpha_host[r];
  }
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

// update + gradient-buffer zeroing fused (native engine: saves one
// fillBuffer launch per accepted round; rejected rounds re-zero g at the
// next dispatch instead)
__global__ void sgd_update_zero_kernel(float* __restrict__ w,
                                       float* __restrict__ g, float gamma_k,
                                       float inv_batch, int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) {
    w[i] -= gamma_k * inv_batch * g[i];
    g[i] = 0.f;
  }
}

__global__ void saga_update_zero_kernel(float* __restrict__ w,
                                        float* __restrict__ g,
                                        float* __restrict__ alpha_bar,
                                        float gamma, float inv_batch,
                                        float inv_N, int d) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < d) {
    const float gi = g[i];
    w[i] -= gamma * (inv_batch * gi + alpha_bar[i]);
    alpha_bar[i] += inv_N * gi;
    g[i] = 0.f;
  }
}

__global__ void saga_commit_kernel(float* __restrict__ alpha,
                                   const int* __restrict__ idx,
                                   const float* __restrict__ e, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) alpha[idx[i]] = e[i];
}

// Host-spill refresh (BASELINE config 5): pull ONLY the round's sampled
// history scalars from the pinned-host master table into the device
// staging table (a_dst[rows[j]] = a_src[rows[j]]). a_src is a pinned host
// pointer — ROCm pinned allocations are device-visible, so the gather is
// ~n_sampled coalesced-issue 4 B reads over the host link instead of the
// whole-table hipMemcpy the round-1 path did. rows/count come from
// scan_rows_kernel with the same Philox key the gradient kernel uses.
__global__ void alpha_gather_kernel(float* __restrict__ a_dst,
                                    const float* __restrict__ a_src,
                                    const int* __restrict__ rows,
                                    const int* __restrict__ count_dev) {
  const int n = *count_dev;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    const int r = rows[i];
    a_dst[r] = a_src[r];
  }
}

// Commit with the staged count read on-device (n_dev = &ctr[1], written by
// the SAGA gradient kernel): removes the native engine's per-accept 4-byte
// D2H sync. Grid is sized for the staging capacity; excess blocks exit on
// the bound check.
__global__ void saga_commit_devn_kernel(float* __restrict__ alpha,
                                        const int* __restrict__ idx,
                                        const float* __restrict__ e,
                                        const int* __restrict__ n_dev) {
  const int n = *n_dev;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) alpha[idx[i]] = e[i];
}

// ---------------------------------------------------------- batched update
// (native engine): ONE kernel applies a whole sweep of accepted gradients —
// elementwise-sequentially, so the result is bit-identical to launching the
// per-round update kernels back to back — zeroes the consumed gradient
// accumulators, and writes the post-batch w into the weight-snapshot
// buffers of the workers being redispatched (the versioned-broadcast copy
// that dispatch() otherwise does with a separate hipMemcpyAsync each).
// Pointer tables (g_tab/wbuf_tab, one slot per worker) live on the device;
// the per-batch worker ids + ASGD step scales ride in the kernel-arg block
// (<= 4 KB), so a batch costs exactly one launch and no H2D staging.
#include "multi_update.h"

__global__ __launch_bounds__(256) void multi_update_kernel(
    float* __restrict__ w, float* const* __restrict__ g_tab,
    float* const* __restrict__ wbuf_tab, float* __restrict__ alpha_bar,
    float gamma, float inv_batch, float inv_N, int d, MultiUpdateArgs a) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < d;
       i += gridDim.x * blockDim.x) {
    float wi = w[i];
    if (a.algo == 0) {
      for (int j = 0; j < a.n; ++j) {
        float* g = g_tab[a.gw[j]];
        wi -= a.scale[j] * g[i];
        g[i] = 0.f;
      }
    } else {
      float ab = alpha_bar[i];
      for (int j = 0; j < a.n; ++j) {
        float* g = g_tab[a.gw[j]];
        const float gi = g[i];
        wi -= gamma * (inv_batch * gi + ab);
        ab += inv_N * gi;
        g[i] = 0.f;
      }
      alpha_bar[i] = ab;
    }
    w[i] = wi;
    for (int j = 0; j < a.m; ++j) wbuf_tab[a.sw[j]][i] = wi;
  }
}

// Fused reduce+update (overlap graph mode): one kernel sums the per-block
// partial slabs AND applies the ASGD update. Each block reads k at entry
// (k is stable for the whole kernel); the last-finishing block (atomic
// ticket) advances k — every other block has already finished, and the
// last block read k at its own start, so the increment races nothing.
__global__ __launch_bounds__(BLOCK) void sgd_reduce_update_kernel(
    const float* __restrict__ g_part, float* __restrict__ w,
    int* __restrict__ k_dev, int* __restrict__ ticket, float gamma,
    float inv_batch, int num_part, int d, int G, int splits) {
  const int k = *k_dev;
  const float gamma_k =
      (float)((double)gamma / sqrt((double)(k / num_part + 1)));
  const int njc = (d + BLOCK - 1) / BLOCK;
  const int jc = blockIdx.x % njc;
  const int sp = blockIdx.x / njc;
  const int j = jc * BLOCK + threadIdx.x;
  if (j < d) {
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
    if (splits == 1) {
      w[j] -= gamma_k * inv_batch * s;
    } else {
      atomicAdd(&w[j], -gamma_k * inv_batch * s);
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int t = atomicAdd(ticket, 1);
    if (t == (int)gridDim.x - 1) {
      *ticket = 0;
      *k_dev = k + 1;
    }
  }
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
                         hipStream_t stream,
                         unsigned long long* done_flag = nullptr,
                         unsigned long long done_val = 0,
                         unsigned long long* done_arr = nullptr) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  const char* np = std::getenv("ASYNCAMD_NO_PIPE");
  const bool pipe_ok = (d % 4 == 0) && (d <= 2048) && !(np && np[0] == '1');
  if (pipe_ok) {
    const int iters = (d + 255) / 256;
    const size_t smem = (size_t)(1 + 4) * (iters * 256) * sizeof(float) +
                        (size_t)QCAP * sizeof(float) * (SAGA ? 3 : 1) +
                        (QCAP + 1) * sizeof(int);
    const char* dps = std::getenv("ASYNCAMD_PIPE_DEPTH");
    const int depth = dps ? std::atoi(dps) : 4;
#define LAUNCH_PIPE_D(IT, DP)                                                \
    hipLaunchKernelGGL((grad_dense_pipe_kernel<XT, SAGA, 256, DP, IT>),      \
                       dim3(grid), dim3(256), smem, stream, X, y, w, g_out,  \
                       g_part, n_out, alpha, idx_out, e_out, pos_ctr,        \
                       k_dev, commit_now, n_rows, d, seed, round_k,          \
                       row_start, thr, take_all, objective, done_flag,       \
                       done_val, done_arr)
#define LAUNCH_PIPE(IT)                                                      \
    do {                                                                     \
      if (depth == 1) LAUNCH_PIPE_D(IT, 1);                                  \
      else if (depth == 2) LAUNCH_PIPE_D(IT, 2);                             \
      else if (depth == 6) LAUNCH_PIPE_D(IT, 6);                             \
      else if (depth == 8) LAUNCH_PIPE_D(IT, 8);                             \
      else LAUNCH_PIPE_D(IT, 4);                                             \
    } while (0)
    switch (iters) {
      case 1: LAUNCH_PIPE(1); break;
      case 2: LAUNCH_PIPE(2); break;
      case 3: LAUNCH_PIPE(3); break;
      case 4: LAUNCH_PIPE(4); break;
      case 5: LAUNCH_PIPE(5); break;
      case 6: LAUNCH_PIPE(6); break;
      case 7: LAUNCH_PIPE(7); break;
      default: LAUNCH_PIPE(8); break;
    }
#undef LAUNCH_PIPE
#undef LAUNCH_PIPE_D
    return;
  }
  const int lpr = pick_lpr(d);
  const size_t smem = (size_t)(1 + 4 * (WAVE / lpr)) * d * sizeof(float);
#define DISPATCH_LPR(L)                                                      \
  hipLaunchKernelGGL((grad_dense_kernel<XT, SAGA, L>), dim3(grid),           \
                     dim3(BLOCK), smem, stream, X, y, w, g_out, g_part,      \
                     n_out, alpha, idx_out, e_out, pos_ctr, k_dev,           \
                     commit_now, n_rows, d, seed, round_k, row_start, thr,   \
                     take_all, objective, done_flag, done_val, done_arr)
  if (lpr == 16) DISPATCH_LPR(16);
  else if (lpr == 32) DISPATCH_LPR(32);
  else DISPATCH_LPR(64);
#undef DISPATCH_LPR
}

extern "C" {

int query_grad_grid(long n_rows) { return grad_grid(n_rows); }

void launch_scan_rows(const float* y, int* rowlist, float* ylist,
                      int* count_dev, const int* scan_round_dev, long n_rows,
                      uint64_t seed, uint32_t round_k, uint64_t row_start,
                      double rate, hipStream_t stream) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  hipLaunchKernelGGL(scan_rows_kernel, dim3(grid), dim3(BLOCK), 0, stream, y,
                     rowlist, ylist, count_dev, scan_round_dev, n_rows, seed,
                     round_k, row_start, thr, take_all);
}

void launch_bump_counter(int* p, hipStream_t stream) {
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, stream, p);
}

void launch_grad_dense_list(const void* X, const float* w, float* g_part,
                            const int* rowlist, const float* ylist,
                            const int* count_dev, long n_rows, int d,
                            int objective, int x_is_bf16,
                            hipStream_t stream) {
  const int grid = grad_grid(n_rows);
  const int iters = (d + 255) / 256;
  const char* dps = std::getenv("ASYNCAMD_PIPE_DEPTH");
  const int depth = dps ? std::atoi(dps) : 4;
  const size_t smem = (size_t)5 * (iters * 256) * sizeof(float) +
                      2048 * (sizeof(float) + sizeof(int));
#define LIST_LAUNCH_D(XT, CAST, IT, DP)                                      \
  hipLaunchKernelGGL((grad_dense_list_kernel<XT, 256, DP, IT>), dim3(grid),  \
                     dim3(256), smem, stream, (CAST)X, w, g_part, rowlist,   \
                     ylist, count_dev, n_rows, d, objective)
#define LIST_LAUNCH(XT, CAST, IT)                                            \
  do {                                                                       \
    if (depth == 2) LIST_LAUNCH_D(XT, CAST, IT, 2);                          \
    else if (depth == 6) LIST_LAUNCH_D(XT, CAST, IT, 6);                     \
    else if (depth == 8) LIST_LAUNCH_D(XT, CAST, IT, 8);                     \
    else LIST_LAUNCH_D(XT, CAST, IT, 4);                                     \
  } while (0)
#define LIST_DISPATCH(XT, CAST)                                              \
  do {                                                                       \
    switch (iters) {                                                         \
      case 1: LIST_LAUNCH(XT, CAST, 1); break;                               \
      case 2: LIST_LAUNCH(XT, CAST, 2); break;                               \
      case 3: LIST_LAUNCH(XT, CAST, 3); break;                               \
      case 4: LIST_LAUNCH(XT, CAST, 4); break;                               \
      case 5: LIST_LAUNCH(XT, CAST, 5); break;                               \
      case 6: LIST_LAUNCH(XT, CAST, 6); break;                               \
      case 7: LIST_LAUNCH(XT, CAST, 7); break;                               \
      default: LIST_LAUNCH(XT, CAST, 8); break;                              \
    }                                                                        \
  } while (0)
  if (x_is_bf16) LIST_DISPATCH(__hip_bfloat16, const __hip_bfloat16*);
  else LIST_DISPATCH(float, const float*);
#undef LIST_DISPATCH
#undef LIST_LAUNCH
#undef LIST_LAUNCH_D
}

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

void launch_grad_dense_flag(
    const void* X, const float* y, const float* w, float* g_out,
    float* g_part, int* n_out, const int* k_dev, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, double rate,
    int objective, int x_is_bf16, hipStream_t stream,
    unsigned long long* done_flag, unsigned long long done_val,
    unsigned long long* done_arr) {
  if (x_is_bf16)
    launch_dense<__hip_bfloat16, false>(
        (const __hip_bfloat16*)X, y, w, g_out, g_part, n_out, nullptr,
        nullptr, nullptr, nullptr, k_dev, 0, n_rows, d, seed, round_k,
        row_start, rate, objective, stream, done_flag, done_val, done_arr);
  else
    launch_dense<float, false>((const float*)X, y, w, g_out, g_part, n_out,
                               nullptr, nullptr, nullptr, nullptr, k_dev, 0,
                               n_rows, d, seed, round_k, row_start, rate,
                               objective, stream, done_flag, done_val,
                               done_arr);
}

void launch_grad_dense_wave(const void* slots_dev, const void* cmd_host,
                            long max_rows, int d, uint64_t seed, double rate,
                            int objective, int x_is_bf16, int saga,
                            hipStream_t stream) {
  const GradWaveCmd* cmd = (const GradWaveCmd*)cmd_host;
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = cmd->n * cmd->bper;
  const int iters = (d + 255) / 256;
  const size_t smem = (size_t)(1 + 4) * (iters * 256) * sizeof(float) +
                      (size_t)QCAP * sizeof(float) * (saga ? 3 : 1) +
                      (QCAP + 1) * sizeof(int);
  const char* dps = std::getenv("ASYNCAMD_PIPE_DEPTH");
  // wave-measured: depth 6 wins at small ITERS (d<=1024, flagship +8%);
  // keep 4 at larger feature dims where the deeper pipeline's register
  // pressure bites (epsilon d=2000). Solo kernels keep 4 everywhere.
  const int depth = dps ? std::atoi(dps) : (iters <= 4 ? 6 : 4);
  (void)max_rows;
#define WAVE_LAUNCH_D(XT, SG, IT, DP)                                        \
  hipLaunchKernelGGL((grad_dense_wave_kernel<XT, SG, 256, DP, IT>),          \
                     dim3(grid), dim3(256), smem, stream,                    \
                     (const GradWaveSlot*)slots_dev, *cmd, d, seed, thr,     \
                     take_all, objective)
#define WAVE_LAUNCH(XT, SG, IT)                                              \
  do {                                                                       \
    if (depth == 1) WAVE_LAUNCH_D(XT, SG, IT, 1);                            \
    else if (depth == 2) WAVE_LAUNCH_D(XT, SG, IT, 2);                       \
    else if (depth == 6) WAVE_LAUNCH_D(XT, SG, IT, 6);                       \
    else if (depth == 8) WAVE_LAUNCH_D(XT, SG, IT, 8);                       \
    else WAVE_LAUNCH_D(XT, SG, IT, 4);                                       \
  } while (0)
#define WAVE_ITERS(XT, SG)                                                   \
  do {                                                                       \
    switch (iters) {                                                         \
      case 1: WAVE_LAUNCH(XT, SG, 1); break;                                 \
      case 2: WAVE_LAUNCH(XT, SG, 2); break;                                 \
      case 3: WAVE_LAUNCH(XT, SG, 3); break;                                 \
      case 4: WAVE_LAUNCH(XT, SG, 4); break;                                 \
      case 5: WAVE_LAUNCH(XT, SG, 5); break;                                 \
      case 6: WAVE_LAUNCH(XT, SG, 6); break;                                 \
      case 7: WAVE_LAUNCH(XT, SG, 7); break;                                 \
      default: WAVE_LAUNCH(XT, SG, 8); break;                                \
    }                                                                        \
  } while (0)
#define WAVE_DISPATCH(XT)                                                    \
  do {                                                                       \
    if (saga) WAVE_ITERS(XT, true);                                          \
    else WAVE_ITERS(XT, false);                                              \
  } while (0)
  if (x_is_bf16) WAVE_DISPATCH(__hip_bfloat16);
  else WAVE_DISPATCH(float);
#undef WAVE_DISPATCH
#undef WAVE_ITERS
#undef WAVE_LAUNCH
#undef WAVE_LAUNCH_D
}

void launch_scan_rows_wave(const void* slots_dev, const void* cmd_host,
                           uint64_t seed, double rate, hipStream_t stream) {
  const GradWaveCmd* cmd = (const GradWaveCmd*)cmd_host;
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  hipLaunchKernelGGL(scan_rows_wave_kernel, dim3(cmd->n * cmd->bper),
                     dim3(BLOCK), 0, stream,
                     (const GradWaveSlot*)slots_dev, *cmd, seed, thr,
                     take_all);
}

void launch_alpha_gather_wave(const void* slots_dev, const void* cmd_host,
                              hipStream_t stream) {
  const GradWaveCmd* cmd = (const GradWaveCmd*)cmd_host;
  hipLaunchKernelGGL(alpha_gather_wave_kernel, dim3(cmd->n * cmd->bper),
                     dim3(BLOCK), 0, stream,
                     (const GradWaveSlot*)slots_dev, *cmd);
}

void launch_saga_grad_dense_flag(
    const void* X, const float* y, const float* w, float* alpha,
    float* g_out, float* g_part, int* n_out, int* idx_out, float* e_out,
    int* pos_ctr, const int* k_dev, int commit_now, long n_rows, int d,
    uint64_t seed, uint32_t round_k, uint64_t row_start, double rate,
    int objective, int x_is_bf16, hipStream_t stream,
    unsigned long long* done_flag, unsigned long long done_val,
    unsigned long long* done_arr) {
  if (x_is_bf16)
    launch_dense<__hip_bfloat16, true>(
        (const __hip_bfloat16*)X, y, w, g_out, g_part, n_out, alpha, idx_out,
        e_out, pos_ctr, k_dev, commit_now, n_rows, d, seed, round_k,
        row_start, rate, objective, stream, done_flag, done_val, done_arr);
  else
    launch_dense<float, true>((const float*)X, y, w, g_out, g_part, n_out,
                              alpha, idx_out, e_out, pos_ctr, k_dev,
                              commit_now, n_rows, d, seed, round_k, row_start,
                              rate, objective, stream, done_flag, done_val,
                              done_arr);
}

// event-free CSR launchers keep their original names as null-flag wrappers
void launch_grad_csr_flag(
    const int* indptr, const int* indices, const void* values,
    const float* y, const float* w, float* g_out, int* n_out,
    const int* k_dev, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, double rate, int objective, int v_is_bf16,
    hipStream_t stream, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr);
void launch_saga_grad_csr_flag(
    const int* indptr, const int* indices, const void* values,
    const float* y, const float* w, float* alpha, float* g_out, int* n_out,
    int* idx_out, float* e_out, int* pos_ctr, const int* k_dev,
    int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, double rate, int objective, int v_is_bf16,
    hipStream_t stream, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr);

void launch_grad_csr(const int* indptr, const int* indices,
                     const void* values, const float* y, const float* w,
                     float* g_out, int* n_out, const int* k_dev, long n_rows,
                     uint64_t seed, uint32_t round_k, uint64_t row_start,
                     double rate, int objective, int v_is_bf16,
                     hipStream_t stream) {
  launch_grad_csr_flag(indptr, indices, values, y, w, g_out, n_out, k_dev,
                       n_rows, seed, round_k, row_start, rate, objective,
                       v_is_bf16, stream, nullptr, 0, nullptr);
}

void launch_saga_grad_csr(const int* indptr, const int* indices,
                          const void* values, const float* y, const float* w,
                          float* alpha, float* g_out, int* n_out,
                          int* idx_out, float* e_out, int* pos_ctr,
                          const int* k_dev, int commit_now, long n_rows,
                          uint64_t seed, uint32_t round_k, uint64_t row_start,
                          double rate, int objective, int v_is_bf16,
                          hipStream_t stream) {
  launch_saga_grad_csr_flag(indptr, indices, values, y, w, alpha, g_out,
                            n_out, idx_out, e_out, pos_ctr, k_dev,
                            commit_now, n_rows, seed, round_k, row_start,
                            rate, objective, v_is_bf16, stream, nullptr, 0,
                            nullptr);
}

void launch_reduce_partials(const float* g_part, float* g_out, int d, int G,
                            int splits, hipStream_t stream) {
  const int njc = (d + BLOCK - 1) / BLOCK;
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(njc * splits), dim3(BLOCK),
                     0, stream, g_part, g_out, d, G, splits);
}

static inline int csr_lpr() {
  const char* s = std::getenv("ASYNCAMD_CSR_LPR");
  const int v = s ? std::atoi(s) : 32;  // measured best on rcv1 shape
  return (v == 8 || v == 16 || v == 32 || v == 64) ? v : 32;
}

void launch_grad_csr_wave(const void* slots_dev, const void* cmd_host,
                          uint64_t seed, double rate, int objective,
                          int v_is_bf16, int saga, hipStream_t stream) {
  const GradWaveCmd* cmd = (const GradWaveCmd*)cmd_host;
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = cmd->n * cmd->bper;
  const int lpr = csr_lpr();
#define CSRW_LAUNCH(VT, SG, L)                                               \
  hipLaunchKernelGGL((grad_csr_wave_kernel<VT, SG, L>), dim3(grid),          \
                     dim3(BLOCK), 0, stream,                                 \
                     (const CsrWaveSlot*)slots_dev, *cmd, seed, thr,         \
                     take_all, objective)
#define CSRW_LPR(VT, SG)                                                     \
  do {                                                                       \
    if (lpr == 8) CSRW_LAUNCH(VT, SG, 8);                                    \
    else if (lpr == 32) CSRW_LAUNCH(VT, SG, 32);                             \
    else if (lpr == 64) CSRW_LAUNCH(VT, SG, 64);                             \
    else CSRW_LAUNCH(VT, SG, 16);                                            \
  } while (0)
#define CSRW_DISPATCH(VT)                                                    \
  do {                                                                       \
    if (saga) CSRW_LPR(VT, true);                                            \
    else CSRW_LPR(VT, false);                                                \
  } while (0)
  if (v_is_bf16) CSRW_DISPATCH(__hip_bfloat16);
  else CSRW_DISPATCH(float);
#undef CSRW_DISPATCH
#undef CSRW_LPR
#undef CSRW_LAUNCH
}

void launch_saga_commit_wave(const void* slots_dev, const void* cmd_host,
                             hipStream_t stream) {
  const CsrCommitCmd* cmd = (const CsrCommitCmd*)cmd_host;
  const int grid = cmd->n * cmd->bper;
  hipLaunchKernelGGL(saga_commit_wave_kernel, dim3(grid), dim3(BLOCK), 0,
                     stream, (const CommitSlot*)slots_dev, *cmd);
}

void launch_grad_csr_flag(
    const int* indptr, const int* indices, const void* values,
    const float* y, const float* w, float* g_out, int* n_out,
    const int* k_dev, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, double rate, int objective, int v_is_bf16,
    hipStream_t stream, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  const int lpr = csr_lpr();
#define CSR_LAUNCH(VT, CAST, L)                                              \
  hipLaunchKernelGGL((grad_csr_kernel<VT, false, L>), dim3(grid),            \
                     dim3(BLOCK), 0, stream, indptr, indices, (CAST)values,  \
                     y, w, g_out, n_out, nullptr, nullptr, nullptr, nullptr, \
                     k_dev, 0, n_rows, seed, round_k, row_start, thr,        \
                     take_all, objective, done_flag, done_val, done_arr)
#define CSR_DISPATCH(VT, CAST)                                               \
  do {                                                                       \
    if (lpr == 8) CSR_LAUNCH(VT, CAST, 8);                                   \
    else if (lpr == 32) CSR_LAUNCH(VT, CAST, 32);                            \
    else if (lpr == 64) CSR_LAUNCH(VT, CAST, 64);                            \
    else CSR_LAUNCH(VT, CAST, 16);                                           \
  } while (0)
  if (v_is_bf16) CSR_DISPATCH(__hip_bfloat16, const __hip_bfloat16*);
  else CSR_DISPATCH(float, const float*);
#undef CSR_DISPATCH
#undef CSR_LAUNCH
}

void launch_saga_grad_csr_flag(
    const int* indptr, const int* indices, const void* values,
    const float* y, const float* w, float* alpha, float* g_out, int* n_out,
    int* idx_out, float* e_out, int* pos_ctr, const int* k_dev,
    int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
    uint64_t row_start, double rate, int objective, int v_is_bf16,
    hipStream_t stream, unsigned long long* done_flag,
    unsigned long long done_val, unsigned long long* done_arr) {
  const uint32_t thr = philox_threshold(rate);
  const int take_all = rate >= 1.0;
  const int grid = grad_grid(n_rows);
  const int lpr = csr_lpr();
#define CSR_SLAUNCH(VT, CAST, L)                                             \
  hipLaunchKernelGGL((grad_csr_kernel<VT, true, L>), dim3(grid),             \
                     dim3(BLOCK), 0, stream, indptr, indices, (CAST)values,  \
                     y, w, g_out, n_out, alpha, idx_out, e_out, pos_ctr,     \
                     k_dev, commit_now, n_rows, seed, round_k, row_start,    \
                     thr, take_all, objective, done_flag, done_val,          \
                     done_arr)
#define CSR_SDISPATCH(VT, CAST)                                              \
  do {                                                                       \
    if (lpr == 8) CSR_SLAUNCH(VT, CAST, 8);                                  \
    else if (lpr == 32) CSR_SLAUNCH(VT, CAST, 32);                           \
    else if (lpr == 64) CSR_SLAUNCH(VT, CAST, 64);                           \
    else CSR_SLAUNCH(VT, CAST, 16);                                          \
  } while (0)
  if (v_is_bf16) CSR_SDISPATCH(__hip_bfloat16, const __hip_bfloat16*);
  else CSR_SDISPATCH(float, const float*);
#undef CSR_SDISPATCH
#undef CSR_SLAUNCH
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

void launch_sgd_update_zero(float* w, float* g, float gamma_k,
                            float inv_batch, int d, hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(sgd_update_zero_kernel, dim3(grid), dim3(256), 0,
                     stream, w, g, gamma_k, inv_batch, d);
}

void launch_saga_update_zero(float* w, float* g, float* alpha_bar,
                             float gamma, float inv_batch, float inv_N,
                             int d, hipStream_t stream) {
  const int grid = (d + 255) / 256;
  hipLaunchKernelGGL(saga_update_zero_kernel, dim3(grid), dim3(256), 0,
                     stream, w, g, alpha_bar, gamma, inv_batch, inv_N, d);
}

void launch_saga_commit(float* alpha, const int* idx, const float* e, int n,
                        hipStream_t stream) {
  if (n <= 0) return;
  const int grid = (n + 255) / 256;
  hipLaunchKernelGGL(saga_commit_kernel, dim3(grid), dim3(256), 0, stream,
                     alpha, idx, e, n);
}

void launch_sgd_reduce_update(const float* g_part, float* w, int* k_dev,
                              int* ticket, float gamma, float inv_batch,
                              int num_part, int d, int G, int splits,
                              hipStream_t stream) {
  const int njc = (d + BLOCK - 1) / BLOCK;
  hipLaunchKernelGGL(sgd_reduce_update_kernel, dim3(njc * splits),
                     dim3(BLOCK), 0, stream, g_part, w, k_dev, ticket, gamma,
                     inv_batch, num_part, d, G, splits);
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

void launch_multi_update(float* w, float* const* g_tab,
                         float* const* wbuf_tab, float* alpha_bar,
                         float gamma, float inv_batch, float inv_N, int d,
                         const MultiUpdateArgs* a, hipStream_t stream) {
  int grid = (d + 255) / 256;
  if (grid > 1024) grid = 1024;
  hipLaunchKernelGGL(multi_update_kernel, dim3(grid), dim3(256), 0, stream,
                     w, g_tab, wbuf_tab, alpha_bar, gamma, inv_batch, inv_N,
                     d, *a);
}

void launch_saga_commit_devn(float* alpha, const int* idx, const float* e,
                             const int* n_dev, int cap, hipStream_t stream) {
  const int grid = (cap + 255) / 256;
  hipLaunchKernelGGL(saga_commit_devn_kernel, dim3(grid > 0 ? grid : 1),
                     dim3(256), 0, stream, alpha, idx, e, n_dev);
}

void launch_alpha_gather(float* a_dst, const float* a_src, const int* rows,
                         const int* count_dev, int cap, hipStream_t stream) {
  const int grid = (cap + 255) / 256;
  hipLaunchKernelGGL(alpha_gather_kernel, dim3(grid > 0 ? grid : 1),
                     dim3(256), 0, stream, a_dst, a_src, rows, count_dev);
}

}  // extern "C"
