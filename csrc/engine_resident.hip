// Device-resident async parameter-server engine: the WHOLE bounded-
// staleness loop (dispatch -> versioned weight snapshot -> Philox-sampled
// gradient -> tau filter -> update -> quorum-gated redispatch, plus the
// straggler model and optVars snapshots) runs inside ONE persistent HIP
// kernel. Zero host involvement between launch and completion.
//
// Motivation (measured, profiles/r02): the host-driven native engine is
// bound by ~6-12 us of HIP API time per update (launch + event per worker
// round) — the GPU itself is >95% idle at the flagship config. Moving the
// control plane on-device removes that wall entirely; the new bound is the
// server block's apply loop (~1-2 us/update of HBM traffic on d=784).
//
// Topology (gfx950: 256 CUs, 64-wide waves):
//   block 0                      = the parameter server (single-writer on w,
//                                  matching the reference's updater thread,
//                                  SparkASGDThread.scala:153-226)
//   blocks 1 + w*G .. w*G+G      = worker group w (one logical partition,
//                                  reference Executor/TaskRunner)
// All blocks are co-resident (grid <= max occupancy, checked at launch);
// communication is device-scope acquire/release atomics on go/done
// counters. Every spin loop checks a realtime deadline and aborts the
// kernel rather than hanging the GPU.
//
// Semantics preserved exactly (same contracts as csrc/engine_native.cpp):
//   * versioned broadcast: the SERVER copies w into the worker's wbuf
//     before setting go (single-writer => the snapshot is a consistent
//     iterate; ASYNCbroadcast.scala:21-27 semantics)
//   * sampling: Philox(seed, k_submit+1, row_start+row) — identical keying
//     to the host engines and utils/philox.py (reference seed+k+1)
//   * tau filter: ASGD staleness<=taw on the arrival clock; ASAGA k-ts<=taw
//     (SparkASGDThread.scala:172 / SparkASAGAThread.scala:191)
//   * ASGD step gamma/sqrt(k/P+1) with Scala int division (:190)
//   * quorum gate floor(P*bucket_ratio) (:233-237), straggler model
//     (:124-141) with on-device calibration from accepted round times
//   * SAGA: worker-resident history, commit-on-accept at next dispatch
//     (ScalarMap merge inside the tau test, SparkASAGAThread.scala:206-208)

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#include "philox.h"

#define RES_BLOCK 256
#define RES_MAXP 64
// control flags are padded one per 128-B cache line: hundreds of idle
// blocks poll these with atomic RMWs, and when all P flags share 2 lines
// the coherent point serializes every poll against every other (measured
// as a ~2x gradient-round slowdown at P=32 vs isolated)
#define RES_CSTRIDE 32

namespace {

struct ResidentArgs {
  float* w;                 // [d] master weights
  // per-worker descriptor table, 8 u64 each:
  //   [0]=X ptr, [1]=y ptr, [2]=wbuf ptr, [3]=g ptr, [4]=n_rows,
  //   [5]=row_start, [6]=alpha ptr (SAGA), [7]=alpha_stage ptr (unused)
  const unsigned long long* desc;
  float* alpha_bar;         // [d] (SAGA)
  float* snap_ring;         // [snap_cap][d] optVars ring (may be null)
  unsigned long long* snap_cycles;  // [snap_cap] timestamps (cycles)
  // control arrays, one slot per worker
  unsigned int* go_round;   // server release-stores; workers acquire-spin
  unsigned int* go_key;     // Philox round key for the round
  unsigned int* go_flags;   // bit0: commit staged SAGA scalars (accept_prev)
  unsigned int* done_round; // worker group release-stores on completion
  unsigned int* done_ctr;   // intra-group arrival counter
  unsigned long long* out;  // results, see OUT_* below
  long long N;
  int d, P, G;
  long long iters;
  float gamma;
  unsigned int thresh;      // Philox Bernoulli threshold (rate)
  unsigned long long seed;
  long long taw;
  int gate;
  int x_is_bf16;
  int algo;                 // 0 asgd, 1 asaga
  int objective;            // 0 lsq, 1 logistic
  double coeff;             // straggler model (0 off, -1 cloud, >0 worker 0)
  long long calib_window;
  double cycles_per_ms;
  unsigned long long deadline_cycles;  // abort watchdog (absolute)
  long long mark_lo, mark_hi;
  long long snap_every, snap_cap;
  float inv_batch, inv_N;
};

enum {
  OUT_K = 0, OUT_APPLIED, OUT_REJECTED, OUT_MAXSTALE, OUT_MARKLO_C,
  OUT_MARKHI_C, OUT_ABORT, OUT_T0_C, OUT_TEND_C, OUT_SNAPN,
  // device-side profiling (cycles / counts)
  OUT_SRV_LOOPS, OUT_SRV_SWEEP_C, OUT_SRV_DISPATCH_C, OUT_W0_ROUNDS,
  OUT_W0_GRAD_C, OUT_W0_SPIN_C,
  // realtime of the last completion processed: on a deadline abort this
  // separates a LIVE run that ran out of wall budget (e.g. tau=0 rejecting
  // everything — the host engines return partial results there too) from a
  // protocol wedge (no completions at all for the final stretch)
  OUT_LAST_PROG_C,
  // fine-grained latency breakdown (cycles / counts)
  OUT_B_C,           // phase-B classify time, work-bearing sweeps only
  OUT_C_C,           // phase-C apply time, work-bearing sweeps only
  OUT_NSWEEP_WORK,   // sweeps that processed >=1 completion
  OUT_W0_WAKE_C,     // go release -> worker-0 round start
  OUT_W0_WAKE_N,
  OUT_W0_DET_C,      // worker-0 done publish -> server classify
  OUT_W0_DET_N,
  OUT_W0_SUB_LAST,   // scratch: server stamps w0 dispatch / w0 stamps done
  OUT_W0_DONE_LAST,
  // abort diagnostics: per-worker server state, packed (see server_block's
  // abort dump); only written when the deadline watchdog fires
  OUT_DUMP, OUT_N = OUT_DUMP + RES_MAXP
};

__device__ __forceinline__ unsigned long long realtime() {
  return __builtin_amdgcn_s_memrealtime();
}

__device__ __forceinline__ unsigned int load_acq(const unsigned int* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_AGENT);
}

// Spin-friendly flag read. Two traps measured on gfx950:
//  * acquire-loads invalidate the XCD L2 on EVERY poll — a chip full of
//    spinning blocks storms the caches (~60x throughput loss);
//  * plain relaxed atomic LOADS can keep hitting a stale local L2 line
//    (no invalidate => a flag write from another XCD is only observed on
//    eviction, ~ms later).
// A relaxed atomic RMW (fetch_add 0) always resolves at the coherent
// point with no cache-wide side effects — the correct poll primitive.
// Follow a positive poll with ONE acquire fence before touching the data
// the flag publishes.
__device__ __forceinline__ unsigned int load_rlx(unsigned int* p) {
  return __hip_atomic_fetch_add(p, 0u, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void acq_fence() {
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
}

__device__ __forceinline__ void store_rel(unsigned int* p, unsigned int v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

// compile-time element conversion: a runtime `x_is_bf16 ?` ternary keeps a
// BRANCH around every load, and branchy loads make the compiler drain
// vmcnt(0) per element (zero memory-level parallelism — the same trap the
// r01 pipe kernel hit; verified in this kernel's ISA)
__device__ __forceinline__ float to_float(float v) { return v; }
__device__ __forceinline__ float to_float(unsigned short u) {
  return bf16_to_f32(u);
}

// ---------------------------------------------------------------- worker

// One worker-group block's share of a round: scan its slice of the shard
// with the round's Philox key, accumulate e_r * x_r of sampled rows into
// its LDS slab, then flush the slab into the worker's global g with
// device-scope atomics. SAGA: also gather alpha, emit new scalars into the
// staging region (g buffer tail), commit previous round's on go_flags bit0.
#define RES_ROWCAP 1024  // sampled-row list per scan window (= window size)
#define RES_MAXNV 32     // register-cached x chunks => d <= 64*32 = 2048

template <typename XT>
__device__ void worker_round(const ResidentArgs& a, int w, int b,
                             unsigned int key, unsigned int flags,
                             float* lds_g) {
  const unsigned long long* D = a.desc + (size_t)w * 8;
  const XT* X = (const XT*)D[0];
  const float* y = (const float*)D[1];
  const float* wbuf = (const float*)D[2];
  float* g = (float*)D[3];
  const long long n_rows = (long long)D[4];
  const long long row_start = (long long)D[5];
  float* alpha = (float*)D[6];      // SAGA only
  const int d = a.d;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  constexpr int WAVES = RES_BLOCK / 64;

  __shared__ int rows_s[RES_ROWCAP];
  __shared__ int qn_s;

  for (int j = tid; j < d; j += RES_BLOCK) lds_g[j] = 0.f;

  // static row partition: block b owns rows [b*slice, min(n, (b+1)*slice)),
  // 4-aligned so the 4-row Philox blocks stay within one owner
  const long long slice = ((n_rows + a.G - 1) / a.G + 3) & ~3ll;
  const long long r0 = (long long)b * slice;
  const long long r1 = (r0 + slice < n_rows) ? r0 + slice : n_rows;

  // windows of WAVES*256 rows: scan (lane = one Philox block of 4 rows,
  // sampled ids appended to the LDS list), then process WAVE-PER-ROW with
  // coalesced loads and the x chunk held in registers between the dot and
  // the rank-1 accumulate (single-lane row processing was ~0.3 ms/row:
  // divergent scalar loads on a dependent FMA chain)
  for (long long cb = r0; cb < r1; cb += (long long)WAVES * 256) {
    __syncthreads();
    if (tid == 0) qn_s = 0;
    __syncthreads();
    const long long base = cb + (long long)wave * 256;
    const long long row4 = base + (long long)lane * 4;
    if (row4 < r1) {
      const uint4 rnd = philox_block4(
          a.seed, key, (unsigned long long)(row_start + row4) / 4);
      const unsigned int rv[4] = {rnd.x, rnd.y, rnd.z, rnd.w};
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const long long r = row4 + q;
        if (r < r1 && rv[q] < a.thresh)
          rows_s[atomicAdd(&qn_s, 1)] = (int)r;
      }
    }
    __syncthreads();
    const int qn = qn_s;
    const int nfull = d >> 6;        // unconditional 64-wide chunks — a
    const int tail = d - (nfull << 6);  // per-element j<d check makes hipcc
                                        // drain vmcnt at every load
    // R rows in flight per wave: the dot pass issues R independent
    // coalesced load streams (hides the ~600-cycle HBM latency that made
    // one-row-at-a-time ~2.5 us/row); the accumulate pass RE-READS x from
    // L1/L2 (cheaper than register-caching the chunks, which made hipcc
    // spill the runtime-indexed array to scratch) and does ONE LDS atomic
    // per chunk for all R rows.
    constexpr int R = 4;
    for (int i0 = wave; i0 < qn; i0 += WAVES * R) {
      const XT* xp[R];
      float p[R];
      int nr = 0;
#pragma unroll
      for (int u = 0; u < R; ++u) {
        p[u] = 0.f;
        const int idx = i0 + u * WAVES;
        if (idx < qn) {
          xp[u] = X + (size_t)rows_s[idx] * d;
          nr = u + 1;
        } else {
          xp[u] = X;  // dummy; masked out of p/scale below
        }
      }
      for (int t = 0; t < nfull; ++t) {
        const int j = t * 64 + lane;
        const float wv = wbuf[j];
#pragma unroll
        for (int u = 0; u < R; ++u) p[u] += to_float(xp[u][j]) * wv;
      }
      if (tail && lane < tail) {
        const int j = nfull * 64 + lane;
        const float wv = wbuf[j];
#pragma unroll
        for (int u = 0; u < R; ++u) p[u] += to_float(xp[u][j]) * wv;
      }
      float scale[R];
#pragma unroll
      for (int u = 0; u < R; ++u) {
        float pu = p[u];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) pu += __shfl_xor(pu, off);
        if (u >= nr) {
          scale[u] = 0.f;
          continue;
        }
        const int r = rows_s[i0 + u * WAVES];
        float e;
        if (a.objective == 1)
          e = 1.f / (1.f + __expf(-pu)) - y[r];
        else
          e = pu - y[r];
        scale[u] = e;
        if (a.algo == 1) {
          // SAGA: corrected gradient (e - alpha_r) * x, new scalar = e
          scale[u] = e - alpha[r];
          if (lane == 0) ((float*)D[7])[r] = e;  // accept-gated staging
        }
      }
      for (int t = 0; t < nfull; ++t) {
        const int j = t * 64 + lane;
        float acc = 0.f;
#pragma unroll
        for (int u = 0; u < R; ++u) acc += scale[u] * to_float(xp[u][j]);
        atomicAdd(&lds_g[j], acc);
      }
      if (tail && lane < tail) {
        const int j = nfull * 64 + lane;
        float acc = 0.f;
#pragma unroll
        for (int u = 0; u < R; ++u) acc += scale[u] * to_float(xp[u][j]);
        atomicAdd(&lds_g[j], acc);
      }
    }
  }
  __syncthreads();
  for (int j = tid; j < d; j += RES_BLOCK)
    if (lds_g[j] != 0.f) atomicAdd(&g[j], lds_g[j]);
  (void)flags;
}

// SAGA commit pass: before computing round `key`, fold the PREVIOUS
// accepted round's staged scalars (identified by re-running its Philox
// mask) into alpha. prev_key == 0 means nothing to commit.
__device__ void commit_pass(const ResidentArgs& a, int w, int b,
                            unsigned int prev_key) {
  const unsigned long long* D = a.desc + (size_t)w * 8;
  const long long n_rows = (long long)D[4];
  const long long row_start = (long long)D[5];
  float* alpha = (float*)D[6];
  const float* stage = (const float*)D[7];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  constexpr int WAVES = RES_BLOCK / 64;
  const long long slice = ((n_rows + a.G - 1) / a.G + 3) & ~3ll;
  const long long r0 = (long long)b * slice;
  const long long r1 = (r0 + slice < n_rows) ? r0 + slice : n_rows;
  for (long long base = r0 + (long long)wave * 256; base < r1;
       base += (long long)WAVES * 256) {
    const long long row4 = base + (long long)lane * 4;
    if (row4 >= r1) continue;
    const uint4 rnd = philox_block4(a.seed, prev_key,
                                    (unsigned long long)(row_start + row4) / 4);
    const unsigned int rv[4] = {rnd.x, rnd.y, rnd.z, rnd.w};
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const long long r = row4 + q;
      if (r < r1 && rv[q] < a.thresh) alpha[r] = stage[r];
    }
  }
}

// ---------------------------------------------------------------- server

// Per-worker server state lives in LDS (P <= RES_MAXP). Thread 0 runs the
// control logic in ONE scalar classify pass per sweep; decisions land in
// LDS op LISTS, and the whole block then executes them BATCHED: one
// elementwise-sequential apply pass over all accepted gradients (identical
// per-element arithmetic order to the one-at-a-time host engines), one
// fan-out copy pass over all dispatch targets, one fence per pass. The
// previous per-worker op loop cost ~3 syncthreads + a fence per worker per
// sweep (~13.7 us/update at P=32, measured via OUT_SRV_* counters).
struct ServerState {
  unsigned int round_no[RES_MAXP];
  unsigned int busy[RES_MAXP];
  unsigned int prev_key[RES_MAXP];     // last accepted round key (SAGA)
  unsigned int last_accept[RES_MAXP];
  int ts[RES_MAXP];                    // arrival clock at dispatch
  long long ksub[RES_MAXP];
  unsigned long long submit_c[RES_MAXP], finish_c[RES_MAXP];
  unsigned long long due_c[RES_MAXP];  // straggler release time (0 = none)
  int straggler_kind[RES_MAXP];
  // persistent pending queue (a gate hold must NOT drop requeued workers;
  // ring, thread-0 mutated only)
  int pendq[RES_MAXP];
  int pq_head, pq_n;
  // ---- batched-sweep op lists (thread 0 writes; block reads after
  // syncthreads) ----
  unsigned int done_snap[RES_MAXP];    // phase-A parallel poll results
  unsigned long long acc_g[RES_MAXP];  // accepted g ptrs, completion order
  float acc_scale[RES_MAXP];
  int nacc;
  unsigned long long rej_g[RES_MAXP];  // rejected g ptrs (zero-only)
  int nrej;
  // apply-pass split points: snapshots and bench marks must observe w at
  // their exact k, so the fused pass breaks there (nseg == 1 in steady
  // state: snap_every is printer_freq-scale, marks fire twice per run)
  int seg_end[RES_MAXP + 1];
  int seg_snap[RES_MAXP + 1];
  int seg_mark[RES_MAXP + 1];          // 0 none / 1 mark_lo / 2 mark_hi
  int nseg;
  unsigned long long dis_wbuf[RES_MAXP];  // batched dispatch fan-out
  int dis_w[RES_MAXP];
  int ndis;
  unsigned long long sweep_c;  // loop-top clock, reused across the sweep
  unsigned long long prof_b;   // phase-B/C profiling scratch (tid 0)
  long long k;
  int clock_;
  long long applied, rejected, max_stale;
  double cul_ms;
  long long cul_n;
  double avg_delay_ms;
  int delay_flag;
};

__device__ __forceinline__ double dev_uniform01(unsigned long long seed,
                                                unsigned int round_k,
                                                unsigned int stream) {
  // counters (0, 0, round, stream) == host uniform01 in the host engines
  uint32_t c0 = 0, c1 = 0, c2 = round_k, c3 = stream;
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint64_t p0 = 0xD2511F53ull * (uint64_t)c0;
    uint64_t p1 = 0xCD9E8D57ull * (uint64_t)c2;
    uint32_t hi0 = (uint32_t)(p0 >> 32), lo0 = (uint32_t)p0;
    uint32_t hi1 = (uint32_t)(p1 >> 32), lo1 = (uint32_t)p1;
    c0 = hi1 ^ c1 ^ k0;
    c1 = lo1;
    c2 = hi0 ^ c3 ^ k1;
    c3 = lo0;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  return c0 / 4294967296.0;
}

__device__ double delay_ms_for_dev(const ResidentArgs& a,
                                   const ServerState& st, int wid,
                                   long long round_k) {
  if (!st.delay_flag || a.coeff == 0.0) return 0.0;
  if (a.coeff != -1.0) {
    if (wid == 0 && a.coeff > 0)
      return round(a.coeff * st.avg_delay_ms);
    return 0.0;
  }
  if (st.straggler_kind[wid] == 2)
    return round((dev_uniform01(a.seed, (unsigned)round_k, wid) * 7.5 + 2.5) *
                 st.avg_delay_ms);
  if (st.straggler_kind[wid] == 1)
    return round((dev_uniform01(a.seed, (unsigned)round_k, wid) + 1.5) *
                 st.avg_delay_ms);
  return 0.0;
}

__device__ void server_block(const ResidentArgs& a) {
  __shared__ ServerState st;
  const int tid = threadIdx.x;
  const int d = a.d;
  if (tid == 0) {
    st.k = 0;
    st.clock_ = 0;
    st.applied = st.rejected = 0;
    st.max_stale = -1;
    st.cul_ms = 0;
    st.cul_n = 0;
    st.avg_delay_ms = 0;
    st.delay_flag = 0;
    st.pq_head = 0;
    st.pq_n = 0;
    for (int i = 0; i < a.P; ++i) {
      st.round_no[i] = 0;
      st.busy[i] = 0;
      st.prev_key[i] = 0;
      st.last_accept[i] = 1;
      st.ts[i] = 0;
      st.ksub[i] = 0;
      st.submit_c[i] = st.finish_c[i] = 0;
      st.due_c[i] = 0;
      st.straggler_kind[i] = 0;
    }
    // reference straggler sets (SparkASGDThread.scala:124-141)
    const int length = (int)llround(0.25 * a.P);
    const int length_normal = (int)llround(0.8 * length);
    const int length_longtail = length - length_normal;
    for (int c = 0; c < length; ++c) {
      const int idx = c * 4;
      if (idx < a.P)
        st.straggler_kind[idx] = (c < length_longtail) ? 2 : 1;
    }
    a.out[OUT_T0_C] = realtime();
  }
  __syncthreads();

  // first dispatch: all workers, gate ignored (reference k==0 path) —
  // batched: one fan-out copy pass, one fence, then all the go stores
  for (int j = tid; j < d; j += RES_BLOCK) {
    const float wv = a.w[j];
    for (int i = 0; i < a.P; ++i)
      ((float*)(a.desc[(size_t)i * 8 + 2]))[j] = wv;
  }
  __threadfence();  // every thread publishes ITS OWN snapshot writes
  __syncthreads();
  if (tid == 0) {
    for (int w = 0; w < a.P; ++w) {
      st.ts[w] = st.clock_;
      st.ksub[w] = st.k;
      st.busy[w] = 1;
      st.submit_c[w] = realtime();
      a.go_key[w * RES_CSTRIDE] = (unsigned int)(st.ksub[w] + 1);
      a.go_flags[w * RES_CSTRIDE] = 0;
      st.round_no[w] += 1;
    }
    __threadfence();
    for (int w = 0; w < a.P; ++w)
      store_rel(&a.go_round[w * RES_CSTRIDE], st.round_no[w]);
  }
  __syncthreads();

  __shared__ int s_done;

  while (true) {
    if (tid == 0) {
      s_done = 0;
      const unsigned long long now_c = realtime();
      st.sweep_c = now_c;  // ONE clock read per sweep: s_memrealtime is a
                           // slow memory-mapped SALU read, and per-worker
                           // reads (submit/finish/due) were ~40% of the
                           // server's serial batch time at P=32
      if (st.k >= a.iters) s_done = 1;
      if (now_c > a.deadline_cycles) {
        a.out[OUT_ABORT] = 1;
        s_done = 1;
        // wedge diagnostics: busy | straggler-hold<<1 | round_no<<8 |
        // pq membership is recoverable host-side from busy+done tensors
        for (int w = 0; w < a.P; ++w)
          a.out[OUT_DUMP + w] =
              (unsigned long long)(st.busy[w] & 1u) |
              ((unsigned long long)(st.due_c[w] != 0) << 1) |
              ((unsigned long long)st.round_no[w] << 8) |
              ((unsigned long long)(st.ksub[w] & 0xFFFF) << 40);
      }
    }
    __syncthreads();
    if (s_done) break;
    const unsigned long long prof_t0 = (tid == 0) ? realtime() : 0;

    // ---- phase A: parallel completion poll (wave 0, one lane per
    // worker — P serialized thread-0 RMWs were ~0.5 us each) ----
    if (tid < a.P)
      st.done_snap[tid] = load_rlx(&a.done_round[tid * RES_CSTRIDE]);
    __syncthreads();

    // ---- phase B: scalar classify (thread 0; builds the op lists) ----
    if (tid == 0) {
      st.prof_b = realtime();
      st.nacc = st.nrej = st.nseg = 0;
      long long kv = st.k;
      long long snapn = (long long)a.out[OUT_SNAPN];
      int seg_open = 0;
      for (int w = 0; w < a.P; ++w) {
        if (!st.busy[w]) continue;
        if (st.due_c[w] != 0) {
          if (st.sweep_c >= st.due_c[w]) {
            // straggler release: back to pending for immediate dispatch
            st.due_c[w] = 0;
            st.busy[w] = 0;
            st.pendq[(st.pq_head + st.pq_n) % RES_MAXP] = w;
            st.pq_n += 1;
          }
          continue;
        }
        if (st.done_snap[w] != st.round_no[w] || kv >= a.iters) continue;
        if (w == 0) {
          a.out[OUT_W0_DET_C] +=
              realtime() - a.out[OUT_W0_DONE_LAST];
          a.out[OUT_W0_DET_N] += 1;
        }
        st.busy[w] = 0;
        st.finish_c[w] = st.sweep_c;
        const int staleness = st.clock_ - st.ts[w];
        st.clock_ += 1;
        if (staleness > st.max_stale) st.max_stale = staleness;
        const bool accept = (a.algo == 1) ? (kv - st.ts[w]) <= a.taw
                                          : staleness <= a.taw;
        st.pendq[(st.pq_head + st.pq_n) % RES_MAXP] = w;
        st.pq_n += 1;
        if (accept) {
          if (kv < a.calib_window) {
            st.cul_ms +=
                (double)(st.sweep_c - st.submit_c[w]) / a.cycles_per_ms;
            st.cul_n += 1;
          }
          const double gamma_k = a.gamma / sqrt((double)(kv / a.P + 1));
          st.acc_g[st.nacc] = a.desc[(size_t)w * 8 + 3];
          st.acc_scale[st.nacc] =
              (a.algo == 1) ? a.gamma : (float)(gamma_k * a.inv_batch);
          st.nacc += 1;
          seg_open += 1;
          st.prev_key[w] = (unsigned int)(st.ksub[w] + 1);
          st.last_accept[w] = 1;
          // optVars snapshot at the pre-increment printer_freq multiple,
          // after applying (reference :195-198); marks at the
          // post-increment step counts — both split the apply pass
          const bool snap = (a.snap_every > 0 && kv % a.snap_every == 0 &&
                             snapn < a.snap_cap);
          if (snap) snapn += 1;
          kv += 1;
          st.applied += 1;
          const int mark =
              (kv == a.mark_lo) ? 1 : ((kv == a.mark_hi) ? 2 : 0);
          if (snap || mark) {
            st.seg_end[st.nseg] = st.nacc;
            st.seg_snap[st.nseg] = snap ? 1 : 0;
            st.seg_mark[st.nseg] = mark;
            st.nseg += 1;
            seg_open = 0;
          }
        } else {
          st.rejected += 1;
          st.last_accept[w] = 0;
          st.rej_g[st.nrej] = a.desc[(size_t)w * 8 + 3];
          st.nrej += 1;
        }
      }
      if (seg_open > 0 || st.nseg == 0) {  // final unflagged segment
        st.seg_end[st.nseg] = st.nacc;
        st.seg_snap[st.nseg] = 0;
        st.seg_mark[st.nseg] = 0;
        st.nseg += 1;
      }
      st.k = kv;
      if (st.nacc + st.nrej > 0) {
        a.out[OUT_LAST_PROG_C] = st.sweep_c;
        a.out[OUT_B_C] += realtime() - st.prof_b;
        a.out[OUT_NSWEEP_WORK] += 1;
        st.prof_b = realtime();  // reused as phase-C start below
      }
    }
    __syncthreads();
    // ONE acquire for all of this sweep's completions, by every thread
    // that will read g (the workers release-published their writes)
    if (st.nacc + st.nrej > 0) acq_fence();

    // ---- phase C: batched apply. Per element j the accepted updates
    // land in completion order — bitwise-identical to the one-at-a-time
    // host engines — but w/alpha_bar are read+written ONCE per pass, and
    // the per-worker syncthreads chains are gone (j->thread mapping is
    // identical across segments, so no barrier between them). ----
    {
      int u0 = 0;
      for (int s = 0; s < st.nseg; ++s) {
        const int u1 = st.seg_end[s];
        if (u1 > u0) {
          if (a.algo == 1) {
            for (int j = tid; j < d; j += RES_BLOCK) {
              float wj = a.w[j], ab = a.alpha_bar[j];
              for (int u = u0; u < u1; ++u) {
                float* g = (float*)st.acc_g[u];
                const float gj = g[j];
                wj -= a.gamma * (a.inv_batch * gj + ab);
                ab += a.inv_N * gj;
                g[j] = 0.f;
              }
              a.w[j] = wj;
              a.alpha_bar[j] = ab;
            }
          } else {
            for (int j = tid; j < d; j += RES_BLOCK) {
              float wj = a.w[j];
              for (int u = u0; u < u1; ++u) {
                float* g = (float*)st.acc_g[u];
                wj -= st.acc_scale[u] * g[j];
                g[j] = 0.f;
              }
              a.w[j] = wj;
            }
          }
          u0 = u1;
        }
        if (st.seg_snap[s]) {
          const long long si = (long long)a.out[OUT_SNAPN];
          float* dst = a.snap_ring + (size_t)si * d;
          for (int j = tid; j < d; j += RES_BLOCK) dst[j] = a.w[j];
          __syncthreads();  // timestamp after the whole copy
          if (tid == 0) {
            a.snap_cycles[si] = realtime();
            a.out[OUT_SNAPN] = (unsigned long long)(si + 1);
          }
          __syncthreads();  // OUT_SNAPN visible before any next segment
        } else if (st.seg_mark[s]) {
          __syncthreads();  // mark time = all threads through the segment
        }
        if (st.seg_mark[s] && tid == 0)
          a.out[(st.seg_mark[s] == 1) ? OUT_MARKLO_C : OUT_MARKHI_C] =
              realtime();
      }
      // rejected gradients: zero only
      for (int r = 0; r < st.nrej; ++r) {
        float* g = (float*)st.rej_g[r];
        for (int j = tid; j < d; j += RES_BLOCK) g[j] = 0.f;
      }
    }

    // ---- phase D: scalar redispatch decisions (quorum gate + straggler
    // model; thread 0 only — the block is still finishing phase C) ----
    if (tid == 0) {
      if (st.nacc + st.nrej > 0) a.out[OUT_C_C] += realtime() - st.prof_b;
      a.out[OUT_SRV_LOOPS] += 1;
      a.out[OUT_SRV_SWEEP_C] += realtime() - prof_t0;
      st.ndis = 0;
      int avail = 0;
      for (int i = 0; i < a.P; ++i)
        if (!st.busy[i]) ++avail;
      // no new rounds once k hit the budget (the host engines stop
      // dispatching there too; an extra tail round would commit one more
      // SAGA staging batch than the native engine — observed as a flaky
      // alpha-pattern mismatch at P=1 before this guard)
      const int n =
          (st.pq_n > 0 && avail >= a.gate && st.k < a.iters) ? st.pq_n : 0;
      for (int i = 0; i < n; ++i) {
        const int w = st.pendq[st.pq_head];
        st.pq_head = (st.pq_head + 1) % RES_MAXP;
        st.pq_n -= 1;
        const double dly = delay_ms_for_dev(a, st, w, st.k);
        if (dly > 0) {
          st.busy[w] = 1;
          st.due_c[w] =
              st.sweep_c + (unsigned long long)(dly * a.cycles_per_ms);
        } else {
          st.dis_w[st.ndis] = w;
          st.dis_wbuf[st.ndis] = a.desc[(size_t)w * 8 + 2];
          st.ndis += 1;
        }
      }
      if (!st.delay_flag && st.k > a.calib_window) {
        if (st.cul_n > 0) st.avg_delay_ms = st.cul_ms / (double)st.cul_n;
        st.delay_flag = 1;
      }
    }
    __syncthreads();
    const unsigned long long prof_t1 = (tid == 0) ? realtime() : 0;

    // ---- phase E: batched dispatch — ONE read of w fans out to every
    // target wbuf (w[j] was written by this same thread in phase C, so
    // no barrier is needed for the data), one fence, then the go stores
    const int ndis = st.ndis;
    if (ndis > 0) {
      for (int j = tid; j < d; j += RES_BLOCK) {
        const float wv = a.w[j];
        for (int u = 0; u < ndis; ++u) ((float*)st.dis_wbuf[u])[j] = wv;
      }
      __threadfence();  // publish this thread's snapshots + g zeroing
      __syncthreads();
      if (tid == 0) {
        const unsigned long long sub_c = realtime();
        for (int u = 0; u < ndis; ++u) {
          const int w = st.dis_w[u];
          st.ts[w] = st.clock_;
          st.ksub[w] = st.k;
          st.busy[w] = 1;
          st.due_c[w] = 0;
          st.submit_c[w] = sub_c;
          a.go_key[w * RES_CSTRIDE] = (unsigned int)(st.ksub[w] + 1);
          // SAGA: bit0 = commit previous accepted round's scalars;
          // bits 1..31 carry the previous key
          a.go_flags[w * RES_CSTRIDE] =
              (a.algo == 1 && st.last_accept[w] && st.prev_key[w])
                  ? ((st.prev_key[w] << 1) | 1u)
                  : 0u;
          st.round_no[w] += 1;
        }
        __threadfence();  // publish go_key/go_flags before the go stores
        for (int u = 0; u < ndis; ++u) {
          if (st.dis_w[u] == 0) a.out[OUT_W0_SUB_LAST] = realtime();
          store_rel(&a.go_round[st.dis_w[u] * RES_CSTRIDE],
                    st.round_no[st.dis_w[u]]);
        }
      }
    }
    if (tid == 0) a.out[OUT_SRV_DISPATCH_C] += realtime() - prof_t1;
    __builtin_amdgcn_s_sleep(8);
  }

  // stop: release workers. SAGA: piggyback a FINAL commit order for
  // workers whose last accepted round was never followed by a dispatch
  // (the native engine's tail dispatch commits these; busy workers
  // already committed at their current dispatch, and straggler-held
  // workers stay uncommitted there too — matched exactly)
  __syncthreads();
  if (tid == 0) {
    for (int w = 0; w < a.P; ++w)
      a.go_flags[w * RES_CSTRIDE] = (a.algo == 1 && !st.busy[w] &&
                                     st.last_accept[w] &&
                       st.prev_key[w])
                          ? ((st.prev_key[w] << 1) | 1u)
                          : 0u;
    __threadfence();
    for (int w = 0; w < a.P; ++w)
      store_rel(&a.go_round[w * RES_CSTRIDE], 0xFFFFFFFFu);
    a.out[OUT_K] = (unsigned long long)st.k;
    a.out[OUT_APPLIED] = (unsigned long long)st.applied;
    a.out[OUT_REJECTED] = (unsigned long long)st.rejected;
    a.out[OUT_MAXSTALE] = (unsigned long long)(st.max_stale + 1);  // -1-safe
    a.out[OUT_TEND_C] = realtime();
  }
}

// ---------------------------------------------------------------- kernel

template <typename XT>
__global__ __launch_bounds__(RES_BLOCK) void resident_engine_kernel(
    ResidentArgs a) {
  if (blockIdx.x == 0) {
    server_block(a);
    return;
  }
  extern __shared__ float lds_g[];
  __shared__ unsigned int s_go[3];  // round, key, flags (LDS broadcast)
  const int w = (blockIdx.x - 1) / a.G;
  const int b = (blockIdx.x - 1) % a.G;
  const int tid = threadIdx.x;
  unsigned int r_local = 0;
  while (true) {
    // ONE thread acquire-spins (65k threads hammering agent-scope acquires
    // would storm the L2s with invalidates); the round record is broadcast
    // through LDS. Throttled and deadline-guarded.
    if (tid == 0) {
      unsigned int r;
      const unsigned long long spin_t0 = realtime();
      int spins = 0;  // exponential backoff: fast reaction to a quick go,
                      // low poll pressure when idle (hundreds of spinning
                      // blocks doing ~1 atomic RMW/us saturate the coherent
                      // fabric and slow every other memory client).
                      // s_sleep needs an immediate, hence the ladder.
      while (true) {
        r = load_rlx(&a.go_round[w * RES_CSTRIDE]);
        if (r != r_local) {
          acq_fence();  // one invalidate, only on a real transition
          break;
        }
        // deadline check amortized: s_memrealtime is slow, and checking
        // it every poll added its latency to the go-reaction time
        if ((spins & 15) == 15 && realtime() > a.deadline_cycles) {
          r = 0xFFFFFFFEu;  // distinct from the server's stop sentinel:
          break;            // a deadline exit must NOT run the final
        }                   // commit (go_flags may be stale)
        if (spins < 4)
          __builtin_amdgcn_s_sleep(2);
        else if (spins < 8)
          __builtin_amdgcn_s_sleep(16);
        else
          __builtin_amdgcn_s_sleep(127);
        ++spins;
      }
      if (w == 0 && b == 0)
        a.out[OUT_W0_SPIN_C] += realtime() - spin_t0;
      s_go[0] = r;
      if (r != 0xFFFFFFFEu) {
        s_go[1] = a.go_key[w * RES_CSTRIDE];
        s_go[2] = a.go_flags[w * RES_CSTRIDE];
      }
    }
    __syncthreads();
    const unsigned int r = s_go[0];
    if (r == 0xFFFFFFFEu) return;  // deadline abort: exit, no commit
    if (r == 0xFFFFFFFFu) {        // stop: final pending SAGA commit
      if (a.algo == 1 && (s_go[2] & 1u)) commit_pass(a, w, b, s_go[2] >> 1);
      return;
    }
    r_local = r;
    const unsigned int key = s_go[1];
    const unsigned int flags = s_go[2];
    __syncthreads();
    const unsigned long long grad_t0 =
        (tid == 0 && w == 0 && b == 0) ? realtime() : 0;
    if (tid == 0 && w == 0 && b == 0) {
      a.out[OUT_W0_WAKE_C] += grad_t0 - a.out[OUT_W0_SUB_LAST];
      a.out[OUT_W0_WAKE_N] += 1;
    }
    if (a.algo == 1 && (flags & 1u)) commit_pass(a, w, b, flags >> 1);
    worker_round<XT>(a, w, b, key, flags, lds_g);
    if (tid == 0 && w == 0 && b == 0) {
      a.out[OUT_W0_GRAD_C] += realtime() - grad_t0;
      a.out[OUT_W0_ROUNDS] += 1;
    }
    // group arrival: every thread fences ITS OWN global writes, then one
    // thread per block counts the arrival; the group's last block
    // publishes done with release semantics
    __threadfence();
    __syncthreads();
    if (tid == 0) {
      const unsigned int arrived =
          __hip_atomic_fetch_add(&a.done_ctr[w * RES_CSTRIDE], 1u,
                                 __ATOMIC_ACQ_REL,
                                 __HIP_MEMORY_SCOPE_AGENT);
      if (arrived + 1 == (unsigned int)a.G) {
        __hip_atomic_store(&a.done_ctr[w * RES_CSTRIDE], 0u,
                           __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        if (w == 0) a.out[OUT_W0_DONE_LAST] = realtime();
        store_rel(&a.done_round[w * RES_CSTRIDE], r_local);
      }
    }
    __syncthreads();
  }
}

// realtime-clock calibration helper: one thread writes s_memrealtime
__global__ void read_realtime_kernel(unsigned long long* out) {
  if (threadIdx.x == 0) *out = __builtin_amdgcn_s_memrealtime();
}

}  // namespace

// ---------------------------------------------------------------- binding

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

#define RHIP_CHECK(x)                                                      \
  do {                                                                     \
    hipError_t _e = (x);                                                   \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("resident engine: ") +          \
                               hipGetErrorString(_e) + " @ " #x);          \
  } while (0)

double wall_s() {
  using clk = std::chrono::steady_clock;
  return std::chrono::duration<double>(clk::now().time_since_epoch()).count();
}

// (cycles_at_t0, cycles_per_ms) of the GPU's constant realtime counter
std::pair<unsigned long long, double> calibrate_realtime(hipStream_t s) {
  unsigned long long* dbuf = nullptr;
  RHIP_CHECK(hipMalloc(&dbuf, 2 * sizeof(unsigned long long)));
  hipLaunchKernelGGL(read_realtime_kernel, dim3(1), dim3(64), 0, s, dbuf);
  RHIP_CHECK(hipStreamSynchronize(s));
  const double w0 = wall_s();
  std::this_thread::sleep_for(std::chrono::milliseconds(60));
  hipLaunchKernelGGL(read_realtime_kernel, dim3(1), dim3(64), 0, s,
                     dbuf + 1);
  RHIP_CHECK(hipStreamSynchronize(s));
  const double w1 = wall_s();
  unsigned long long c[2];
  RHIP_CHECK(hipMemcpy(c, dbuf, sizeof(c), hipMemcpyDeviceToHost));
  RHIP_CHECK(hipFree(dbuf));
  const double cpm = (double)(c[1] - c[0]) / ((w1 - w0) * 1000.0);
  return {c[1], cpm};
}

}  // namespace

void register_resident_engine(py::module_& m) {
  m.def("resident_run", [](py::dict c) {
    ResidentArgs a{};
    a.w = (float*)py::cast<uintptr_t>(c["w"]);
    a.desc = (const unsigned long long*)py::cast<uintptr_t>(c["desc"]);
    a.alpha_bar = (float*)py::cast<uintptr_t>(c["alpha_bar"]);
    a.snap_ring = (float*)py::cast<uintptr_t>(c["snap_ring"]);
    a.snap_cycles =
        (unsigned long long*)py::cast<uintptr_t>(c["snap_cycles"]);
    a.go_round = (unsigned int*)py::cast<uintptr_t>(c["go_round"]);
    a.go_key = (unsigned int*)py::cast<uintptr_t>(c["go_key"]);
    a.go_flags = (unsigned int*)py::cast<uintptr_t>(c["go_flags"]);
    a.done_round = (unsigned int*)py::cast<uintptr_t>(c["done_round"]);
    a.done_ctr = (unsigned int*)py::cast<uintptr_t>(c["done_ctr"]);
    a.out = (unsigned long long*)py::cast<uintptr_t>(c["out"]);
    a.N = py::cast<long long>(c["N"]);
    a.d = py::cast<int>(c["d"]);
    a.P = py::cast<int>(c["P"]);
    a.G = py::cast<int>(c["G"]);
    a.iters = py::cast<long long>(c["iters"]);
    a.gamma = (float)py::cast<double>(c["gamma"]);
    const double rate = py::cast<double>(c["rate"]);
    if (rate >= 1.0)
      throw std::runtime_error("resident engine: rate >= 1 unsupported");
    a.thresh = philox_threshold(rate);
    a.seed = py::cast<uint64_t>(c["seed"]);
    a.taw = py::cast<long long>(c["taw"]);
    a.gate = py::cast<int>(c["gate"]);
    a.x_is_bf16 = py::cast<int>(c["x_is_bf16"]);
    a.algo = py::cast<int>(c["algo"]);
    a.objective = py::cast<int>(c["objective"]);
    a.coeff = py::cast<double>(c["coeff"]);
    a.calib_window = py::cast<long long>(c["calib_window"]);
    a.mark_lo = py::cast<long long>(c["mark_lo"]);
    a.mark_hi = py::cast<long long>(c["mark_hi"]);
    a.snap_every = py::cast<long long>(c["snap_every"]);
    a.snap_cap = py::cast<long long>(c["snap_cap"]);
    a.inv_batch = (float)py::cast<double>(c["inv_batch"]);
    a.inv_N = (float)py::cast<double>(c["inv_N"]);
    const double max_wall_s = py::cast<double>(c["max_wall_s"]);
    if (a.P < 1 || a.P > RES_MAXP)
      throw std::runtime_error("resident engine: P must be in [1, 64]");

    py::gil_scoped_release rel;
    hipStream_t stream;
    RHIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    auto [c_now, cpm] = calibrate_realtime(stream);
    a.cycles_per_ms = cpm;
    a.deadline_cycles =
        c_now + (unsigned long long)(max_wall_s * 1000.0 * cpm);

    const int grid = 1 + a.P * a.G;
    const size_t lds = (size_t)a.d * sizeof(float);
    int nblk = 0;
    if (a.x_is_bf16)
      RHIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &nblk, (const void*)resident_engine_kernel<unsigned short>,
          RES_BLOCK, lds));
    else
      RHIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &nblk, (const void*)resident_engine_kernel<float>, RES_BLOCK,
          lds));
    hipDeviceProp_t prop;
    RHIP_CHECK(hipGetDeviceProperties(&prop, 0));
    const int max_resident = nblk * prop.multiProcessorCount;
    if (grid > max_resident)
      throw std::runtime_error(
          "resident engine: grid " + std::to_string(grid) +
          " exceeds co-resident capacity " + std::to_string(max_resident) +
          " (persistent kernel would deadlock); lower P or G");
    if (a.x_is_bf16)
      hipLaunchKernelGGL(resident_engine_kernel<unsigned short>, dim3(grid),
                         dim3(RES_BLOCK), lds, stream, a);
    else
      hipLaunchKernelGGL(resident_engine_kernel<float>, dim3(grid),
                         dim3(RES_BLOCK), lds, stream, a);
    RHIP_CHECK(hipGetLastError());
    // host-side watchdog: poll with a margin past the device deadline
    const double host_deadline = wall_s() + max_wall_s + 10.0;
    while (true) {
      const hipError_t q = hipStreamQuery(stream);
      if (q == hipSuccess) break;
      if (q != hipErrorNotReady) RHIP_CHECK(q);
      if (wall_s() > host_deadline) {
        (void)hipStreamDestroy(stream);
        throw std::runtime_error(
            "resident engine: kernel exceeded host deadline (device "
            "watchdog failed to fire)");
      }
      std::this_thread::sleep_for(std::chrono::microseconds(200));
    }
    RHIP_CHECK(hipStreamDestroy(stream));

    unsigned long long out_h[OUT_N];
    RHIP_CHECK(hipMemcpy(out_h, (const void*)a.out, sizeof(out_h),
                         hipMemcpyDeviceToHost));
    py::gil_scoped_acquire acq;
    py::dict r;
    r["k"] = (long long)out_h[OUT_K];
    r["applied"] = (long long)out_h[OUT_APPLIED];
    r["rejected"] = (long long)out_h[OUT_REJECTED];
    r["max_staleness"] = (long long)out_h[OUT_MAXSTALE] - 1;
    r["aborted"] = (bool)out_h[OUT_ABORT];
    const double t0c = (double)out_h[OUT_T0_C];
    r["elapsed_ms"] = ((double)out_h[OUT_TEND_C] - t0c) / cpm;
    r["mark_lo_t"] = out_h[OUT_MARKLO_C]
                         ? ((double)out_h[OUT_MARKLO_C] - t0c) / cpm / 1e3
                         : 0.0;
    r["mark_hi_t"] = out_h[OUT_MARKHI_C]
                         ? ((double)out_h[OUT_MARKHI_C] - t0c) / cpm / 1e3
                         : 0.0;
    r["snap_n"] = (long long)out_h[OUT_SNAPN];
    r["cycles_per_ms"] = cpm;
    r["t0_cycles"] = (unsigned long long)out_h[OUT_T0_C];
    // device-side profiling (where the cycles went)
    r["srv_loops"] = (long long)out_h[OUT_SRV_LOOPS];
    r["srv_sweep_ms"] = (double)out_h[OUT_SRV_SWEEP_C] / cpm;
    r["srv_dispatch_ms"] = (double)out_h[OUT_SRV_DISPATCH_C] / cpm;
    r["w0_rounds"] = (long long)out_h[OUT_W0_ROUNDS];
    r["w0_grad_ms"] = (double)out_h[OUT_W0_GRAD_C] / cpm;
    r["w0_spin_ms"] = (double)out_h[OUT_W0_SPIN_C] / cpm;
    r["last_progress_ms"] =
        out_h[OUT_LAST_PROG_C]
            ? ((double)out_h[OUT_LAST_PROG_C] - t0c) / cpm
            : -1.0;
    r["srv_classify_ms"] = (double)out_h[OUT_B_C] / cpm;
    r["srv_apply_ms"] = (double)out_h[OUT_C_C] / cpm;
    r["srv_work_sweeps"] = (long long)out_h[OUT_NSWEEP_WORK];
    r["w0_wake_ms"] = (double)out_h[OUT_W0_WAKE_C] / cpm;
    r["w0_wake_n"] = (long long)out_h[OUT_W0_WAKE_N];
    r["w0_detect_ms"] = (double)out_h[OUT_W0_DET_C] / cpm;
    r["w0_detect_n"] = (long long)out_h[OUT_W0_DET_N];
    return r;
  });
}
