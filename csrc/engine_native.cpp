// Native async parameter-server engine (single GPU, multi-worker).
//
// The MI355X-native replacement for the reference's driver main loop +
// updater thread + executor pool (SparkASGDThread.scala:153-345,
// Executor.scala TaskRunner) for co-located workers: ONE host thread drives
// an event loop over per-worker HIP streams —
//
//   dispatch:    each quorum WAVE launches as ONE kernel
//                (grad_dense_wave_kernel / grad_csr_wave_kernel over a
//                device slot table, SAGA commits batched into one
//                commit+staging-reset kernel, spill refresh as
//                scan/gather wave kernels) on a rotating stream pool; the
//                batched update kernel writes the redispatching workers'
//                weight snapshots (versioned-broadcast semantic), so the
//                whole wave costs a handful of launches instead of 2-5
//                per worker. First dispatches and straggler releases use
//                the per-worker singleton path;
//   poll:        EVENT-FREE — the grad kernel's last block publishes the
//                round serial to a pinned-host line (publish_done,
//                kernels.hip) and the host polls plain memory;
//   accept:      tau filter; accepted gradients accumulate into ONE
//                elementwise-sequential multi-update kernel per sweep;
//                requeue; quorum-gated redispatch;
//   delay:       the reference's straggler model (cloud long-tail / coeff),
//                implemented as host-side due-times on the dispatch queue —
//                no thread sleeps, no locks, no GIL (released for the whole
//                run).
//
// This is the same control plane as engine/local.py with the thread
// handoffs (~100 us each under the GIL) replaced by sub-us flag reads.
// Measured evolution on the flagship config (profiles/r02_wave_dispatch.md):
// 39.9k updates/s (r01, per-round launches + events) -> 87.2k (batched
// updates) -> 103k (event-free) -> 188k (wave dispatch, interleaved
// mapping, wave-aware grid sizing). The RCCL multi-GPU server reuses this
// loop shape with peers instead of streams (csrc/server_dist.cpp).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <hip/hip_runtime.h>

#include "grad_wave.h"
#include "multi_update.h"

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <deque>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

extern "C" {
void launch_grad_dense(const void*, const float*, const float*, float*,
                       float*, int*, const int*, long, int, uint64_t,
                       uint32_t, uint64_t, double, int, int, hipStream_t);
void launch_saga_grad_dense(const void*, const float*, const float*, float*,
                            float*, float*, int*, int*, float*, int*,
                            const int*, int, long, int, uint64_t, uint32_t,
                            uint64_t, double, int, int, hipStream_t);
void launch_grad_csr(const int*, const int*, const void*, const float*,
                     const float*, float*, int*, const int*, long, uint64_t,
                     uint32_t, uint64_t, double, int, int, hipStream_t);
void launch_saga_grad_csr(const int*, const int*, const void*, const float*,
                          const float*, float*, float*, int*, int*, float*,
                          int*, const int*, int, long, uint64_t, uint32_t,
                          uint64_t, double, int, int, hipStream_t);
int query_grad_grid(long);
void launch_grad_csr_wave(const void*, const void*, uint64_t, double, int,
                          int, int, hipStream_t);
void launch_saga_commit_wave(const void*, const void*, hipStream_t);
void launch_scan_rows_wave(const void*, const void*, uint64_t, double,
                           hipStream_t);
void launch_alpha_gather_wave(const void*, const void*, hipStream_t);
void launch_grad_dense_wave(const void*, const void*, long, int, uint64_t,
                            double, int, int, int, hipStream_t);
void launch_grad_dense_flag(const void*, const float*, const float*,
                            float*, float*, int*, const int*, long, int,
                            uint64_t, uint32_t, uint64_t, double, int, int,
                            hipStream_t, unsigned long long*,
                            unsigned long long, unsigned long long*);
void launch_saga_grad_dense_flag(const void*, const float*, const float*,
                                 float*, float*, float*, int*, int*, float*,
                                 int*, const int*, int, long, int, uint64_t,
                                 uint32_t, uint64_t, double, int, int,
                                 hipStream_t, unsigned long long*,
                                 unsigned long long, unsigned long long*);
void launch_grad_csr_flag(const int*, const int*, const void*, const float*,
                          const float*, float*, int*, const int*, long,
                          uint64_t, uint32_t, uint64_t, double, int, int,
                          hipStream_t, unsigned long long*,
                          unsigned long long, unsigned long long*);
void launch_saga_grad_csr_flag(const int*, const int*, const void*,
                               const float*, const float*, float*, float*,
                               int*, int*, float*, int*, const int*, int,
                               long, uint64_t, uint32_t, uint64_t, double,
                               int, int, hipStream_t, unsigned long long*,
                               unsigned long long, unsigned long long*);
void launch_sgd_update(float*, const float*, float, float, int, hipStream_t);
void launch_saga_update(float*, const float*, float*, float, float, float,
                        int, hipStream_t);
void launch_saga_commit(float*, const int*, const float*, int, hipStream_t);
void launch_sgd_update_zero(float*, float*, float, float, int, hipStream_t);
void launch_saga_update_zero(float*, float*, float*, float, float, float,
                             int, hipStream_t);
void launch_multi_update(float*, float* const*, float* const*, float*, float,
                         float, float, int, const MultiUpdateArgs*,
                         hipStream_t);
void launch_saga_commit_devn(float*, const int*, const float*, const int*,
                             int, hipStream_t);
void launch_scan_rows(const float*, int*, float*, int*, const int*, long,
                      uint64_t, uint32_t, uint64_t, double, hipStream_t);
void launch_alpha_gather(float*, const float*, const int*, const int*, int,
                         hipStream_t);
}

namespace {

#define HIP_CHECK(x)                                                       \
  do {                                                                     \
    hipError_t _e = (x);                                                   \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("native engine: ") +            \
                               hipGetErrorString(_e) + " @ " #x);          \
  } while (0)

// Host Philox (must match csrc/philox.h / utils/philox.py) for the
// reproducible straggler draws (uniform01 counters (i,0,round,stream)).
inline uint32_t philox_host_x0(uint64_t seed, uint32_t c0, uint32_t c1,
                               uint32_t c2, uint32_t c3) {
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
  for (int r = 0; r < 10; ++r) {
    uint64_t p0 = 0xD2511F53ull * (uint64_t)c0;
    uint64_t p1 = 0xCD9E8D57ull * (uint64_t)c2;
    uint32_t hi0 = (uint32_t)(p0 >> 32), lo0 = (uint32_t)p0;
    uint32_t hi1 = (uint32_t)(p1 >> 32), lo1 = (uint32_t)p1;
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  return c0;
}

inline double uniform01_host(uint64_t seed, uint32_t round_k,
                             uint32_t stream) {
  return philox_host_x0(seed, 0u, 0u, round_k, stream) / 4294967296.0;
}

struct WorkerBuf {
  // device pointers supplied by Python (torch tensors kept alive there)
  uintptr_t X = 0, indptr = 0, indices = 0, values = 0, y = 0;
  uintptr_t alpha = 0, idx_out = 0, e_out = 0;  // SAGA staging
  // host-spill mode (BASELINE config 5): alpha_host = pinned master table
  // (device-visible); alpha above becomes the device staging table that
  // scan_rows + alpha_gather refresh at the round's sampled rows
  uintptr_t alpha_host = 0, srows = 0, sylist = 0, scnt = 0;
  uintptr_t wbuf = 0, g = 0, ctr = 0;           // snapshot, grad, counters
  long n_rows = 0, row_start = 0;
  bool sparse = false;
  int x_is_bf16 = 0;
  // runtime
  hipStream_t stream = nullptr;
  // event-free completion: the grad kernel's last block release-stores
  // round_serial into pinned host memory (publish_done in kernels.hip);
  // the host event loop polls plain memory instead of hipEventQuery
  volatile unsigned long long* done_flag = nullptr;  // pinned host, 1 line
  unsigned long long* done_arr = nullptr;  // device arrival ctr (monotonic)
  unsigned long long round_serial = 0;
  bool busy = false;
  // a round is actually executing on the GPU. Distinct from busy: a
  // straggler-DELAYED worker is busy (occupied) but not in flight, and its
  // done flag still holds the previous round's serial — polling it would
  // double-book that round (latent in the event-based loop too, where the
  // completed event also stayed signaled; made explicit here)
  bool in_flight = false;
  int ts = 0;        // arrival clock at dispatch
  long k_submit = 0; // round index at dispatch
  bool pending_commit = false;
  int saga_cap = 0;  // staging capacity (commit grid bound, SAGA only)
  double submit_t = 0, finish_t = 0, waiting_ms = 0;
  bool g_dirty = false;   // gradient buffer holds a rejected round's sums
  long tasks = 0;
};

struct EngineCfg {
  long N = 0;
  int d = 0, P = 1;
  long iters = 0;
  double gamma = 0.01, rate = 0.01, bucket_ratio = 0.7;
  long taw = 1 << 30;
  uint64_t seed = 42;
  int algo = 0;       // 0 asgd, 1 asaga
  int objective = 0;  // 0 lsq, 1 logistic
  double coeff = 0.0; // delay model
  long calib_window = 0;
  long mark_lo = -1, mark_hi = -1;
  double max_wall_s = 3600.0;
  double stall_s = 60.0;     // watchdog: abort if no completion for this long
  long snap_every = 0;       // optVars cadence (reference printer_freq)
  uintptr_t snap_ring = 0;   // [snap_cap][d] device ring
  long snap_cap = 0;
};

double now_s() {
  using clk = std::chrono::steady_clock;
  return std::chrono::duration<double>(clk::now().time_since_epoch()).count();
}

struct NativeEngine {
  EngineCfg cfg;
  std::vector<WorkerBuf> ws;
  uintptr_t w = 0, alpha_bar = 0;
  hipStream_t sstream = nullptr;  // server stream
  hipEvent_t update_ev = nullptr;
  // server state
  long k = 0;
  int clock = 0;  // arrival clock (ASYNCcontext.CurrentTime)
  std::deque<int> pendingq;
  std::deque<std::pair<double, int>> delayed;  // (due time, worker)
  long applied = 0, rejected = 0;
  long max_staleness_seen = -1;
  // delay calibration (reference :177-186,:247-252)
  double cul_time_ms = 0;
  long cul_count = 0;
  double avg_delay_ms = 0;
  bool delay_flag = false;
  // straggler sets (reference :124-141)
  std::vector<int> straggler_kind;  // 0 none, 1 normal, 2 longtail
  double mark_lo_t = 0, mark_hi_t = 0;
  std::vector<double> snap_ms;  // host stamps for the optVars snapshots
  double run_t0 = 0;
  // batched-update machinery
  float** g_tab_dev = nullptr;     // device table: worker id -> g pointer
  volatile unsigned long long* flags_host = nullptr;
  unsigned long long* arr_dev = nullptr;
  // wave dispatch (dense ASGD): per-worker invariant table + stream
  GradWaveSlot* slots_dev = nullptr;
  CsrWaveSlot* csr_slots_dev = nullptr;
  CommitSlot* commit_slots_dev = nullptr;
  bool wave_spill = false;
  // pool of wave streams: consecutive waves overlap like the per-worker
  // streams did (same-worker overlap is impossible — in_flight guard)
  static constexpr int NWSTREAM = 8;
  hipStream_t wstreams[NWSTREAM] = {};
  unsigned wave_rr = 0;
  int wave_interleave = 0;
  hipStream_t cur_wst = nullptr;
  int wave_bper = 0;
  long wave_max_rows = 0;
  bool wave_ok = false;
  float** wbuf_tab_dev = nullptr;  // device table: worker id -> wbuf pointer
  double inv_batch = 0, inv_N = 0;

  void init_stragglers() {
    straggler_kind.assign(cfg.P, 0);
    const int length = (int)std::lround(0.25 * cfg.P);
    const int length_normal = (int)std::lround(0.8 * length);
    const int length_longtail = length - length_normal;
    for (int c = 0; c < length; ++c) {
      const int idx = c * 4;
      if (idx < cfg.P) straggler_kind[idx] = (c < length_longtail) ? 2 : 1;
    }
  }

  double delay_ms_for(int wid, long round_k) const {
    if (!delay_flag || cfg.coeff == 0.0) return 0.0;
    if (cfg.coeff != -1.0) {
      if (wid == 0 && cfg.coeff > 0) return std::round(cfg.coeff * avg_delay_ms);
      return 0.0;
    }
    if (straggler_kind[wid] == 2) {
      const double u = uniform01_host(cfg.seed, (uint32_t)round_k, wid);
      return std::round((u * 7.5 + 2.5) * avg_delay_ms);
    }
    if (straggler_kind[wid] == 1) {
      const double u = uniform01_host(cfg.seed, (uint32_t)round_k, wid);
      return std::round((u + 1.5) * avg_delay_ms);
    }
    return 0.0;
  }

  int gate() const {
    return std::max(1, (int)std::floor(cfg.P * cfg.bucket_ratio));
  }

  int available() const {
    int n = 0;
    for (auto& wk : ws)
      if (!wk.busy) ++n;
    return n;
  }

  void launch_grad(WorkerBuf& wk, long round_key) {
    // g is zeroed by the fused update of the worker's ACCEPTED previous
    // round; only a rejected round leaves it dirty. ASGD never reads the
    // sampled count, so its ctr memset is skipped.
    if (wk.g_dirty) {
      HIP_CHECK(hipMemsetAsync((void*)wk.g, 0, (size_t)cfg.d * 4, wk.stream));
      wk.g_dirty = false;
    }
    if (cfg.algo == 1)
      HIP_CHECK(hipMemsetAsync((void*)wk.ctr, 0, 8, wk.stream));
    if (cfg.algo == 1 && wk.alpha_host) {
      // spill refresh: recompute the round's Philox row set on-device and
      // gather ONLY those entries from the pinned master into the staging
      // table the gradient kernel reads (ordered after this worker's
      // commit scatter on the same stream)
      HIP_CHECK(hipMemsetAsync((void*)wk.scnt, 0, 4, wk.stream));
      launch_scan_rows((const float*)wk.y, (int*)wk.srows,
                       (float*)wk.sylist, (int*)wk.scnt, nullptr, wk.n_rows,
                       cfg.seed, (uint32_t)round_key, (uint64_t)wk.row_start,
                       cfg.rate, wk.stream);
      launch_alpha_gather((float*)wk.alpha, (const float*)wk.alpha_host,
                          (const int*)wk.srows, (const int*)wk.scnt,
                          wk.saga_cap, wk.stream);
      HIP_CHECK(hipGetLastError());
    }
    if (cfg.algo == 1) {
      if (wk.sparse)
        launch_saga_grad_csr_flag(
            (const int*)wk.indptr, (const int*)wk.indices,
            (const void*)wk.values, (const float*)wk.y,
            (const float*)wk.wbuf, (float*)wk.alpha, (float*)wk.g,
            (int*)wk.ctr, (int*)wk.idx_out, (float*)wk.e_out,
            (int*)(wk.ctr + 4), nullptr, 0, wk.n_rows, cfg.seed,
            (uint32_t)round_key, (uint64_t)wk.row_start, cfg.rate,
            cfg.objective, wk.x_is_bf16, wk.stream,
            (unsigned long long*)wk.done_flag, wk.round_serial,
            wk.done_arr);
      else
        launch_saga_grad_dense_flag(
            (const void*)wk.X, (const float*)wk.y, (const float*)wk.wbuf,
            (float*)wk.alpha, (float*)wk.g, nullptr, (int*)wk.ctr,
            (int*)wk.idx_out, (float*)wk.e_out, (int*)(wk.ctr + 4), nullptr,
            0, wk.n_rows, cfg.d, cfg.seed, (uint32_t)round_key,
            (uint64_t)wk.row_start, cfg.rate, cfg.objective, wk.x_is_bf16,
            wk.stream, (unsigned long long*)wk.done_flag, wk.round_serial,
            wk.done_arr);
    } else {
      if (wk.sparse)
        launch_grad_csr_flag(
            (const int*)wk.indptr, (const int*)wk.indices,
            (const void*)wk.values, (const float*)wk.y,
            (const float*)wk.wbuf, (float*)wk.g, (int*)wk.ctr, nullptr,
            wk.n_rows, cfg.seed, (uint32_t)round_key,
            (uint64_t)wk.row_start, cfg.rate, cfg.objective, wk.x_is_bf16,
            wk.stream, (unsigned long long*)wk.done_flag, wk.round_serial,
            wk.done_arr);
      else
        launch_grad_dense_flag(
            (const void*)wk.X, (const float*)wk.y, (const float*)wk.wbuf,
            (float*)wk.g, nullptr, (int*)wk.ctr, nullptr, wk.n_rows, cfg.d,
            cfg.seed, (uint32_t)round_key, (uint64_t)wk.row_start, cfg.rate,
            cfg.objective, wk.x_is_bf16, wk.stream,
            (unsigned long long*)wk.done_flag, wk.round_serial,
            wk.done_arr);
    }
    HIP_CHECK(hipGetLastError());
  }

  // Common dispatch tail: SAGA commit of the previous accepted round (count
  // read on-device — no D2H sync), waiting-time bookkeeping, grad launch
  // ordered after the latest applied update, completion event.
  // ``copy_w``: true -> snapshot w with a hipMemcpyAsync on the worker
  // stream (first dispatch / delayed release); false -> the batched update
  // kernel already wrote this worker's wbuf on sstream (the common path).
  void dispatch_impl(int wid, double t_now, bool copy_w) {
    WorkerBuf& wk = ws[wid];
    // accept-gated SAGA history commit from the worker's previous round
    // (spill mode: scatter straight into the pinned master table)
    if (cfg.algo == 1 && wk.pending_commit) {
      if (commit_slots_dev) {
        // wave mode: the previous grad ran on a wave stream, so the
        // staging must be read with the RMW-load commit kernel (normal
        // loads could hit a stale L2 line across the stream boundary)
        CsrCommitCmd cc;
        cc.n = 1;
        cc.bper = 16;
        cc.wid[0] = wid;
        cc.do_commit[0] = 1;
        launch_saga_commit_wave(commit_slots_dev, &cc, wk.stream);
      } else {
        float* dst = (float*)(wk.alpha_host ? wk.alpha_host : wk.alpha);
        launch_saga_commit_devn(dst, (const int*)wk.idx_out,
                                (const float*)wk.e_out,
                                (const int*)(wk.ctr + 4), wk.saga_cap,
                                wk.stream);
      }
      HIP_CHECK(hipGetLastError());
    }
    wk.pending_commit = false;
    if (wk.finish_t == 0) wk.finish_t = t_now;  // first dispatch: no wait
    wk.waiting_ms += (t_now - wk.finish_t) * 1000.0;
    wk.submit_t = t_now;
    wk.busy = true;
    wk.ts = clock;
    wk.k_submit = k;
    // versioned weights: the worker computes against the w it was
    // dispatched with (ASYNCbroadcast semantics) — ordered after the
    // latest applied update either way
    HIP_CHECK(hipStreamWaitEvent(wk.stream, update_ev, 0));
    if (copy_w)
      HIP_CHECK(hipMemcpyAsync((void*)wk.wbuf, (const void*)w,
                               (size_t)cfg.d * 4, hipMemcpyDeviceToDevice,
                               wk.stream));
    if (wk.in_flight)
      throw std::runtime_error(
          "native engine invariant: dispatch while a round is in flight "
          "(worker " + std::to_string(wid) + ")");
    wk.round_serial += 1;  // the grad kernel's last block publishes this
    wk.in_flight = true;
    launch_grad(wk, wk.k_submit + 1);  // reference seed+k+1
    // no hipEventRecord: completion = *done_flag == round_serial (plain
    // pinned-memory read, ~0 host cost vs ~1-2 us per hipEventQuery)
  }

  void dispatch(int wid, double t_now) { dispatch_impl(wid, t_now, true); }

  // Wave-path bookkeeping: everything dispatch_impl does EXCEPT the grad
  // launch (the whole quorum wave launches as ONE grad_dense_wave_kernel).
  // ASGD-only: no SAGA commit to chain.
  void dispatch_book(int wid, double t_now) {
    WorkerBuf& wk = ws[wid];
    if (wk.finish_t == 0) wk.finish_t = t_now;  // first dispatch: no wait
    wk.waiting_ms += (t_now - wk.finish_t) * 1000.0;
    wk.submit_t = t_now;
    wk.busy = true;
    wk.ts = clock;
    wk.k_submit = k;
    if (wk.g_dirty) {  // rejected round left sums; zero on the wave stream
      HIP_CHECK(hipMemsetAsync((void*)wk.g, 0, (size_t)cfg.d * 4,
                               cur_wst));
      wk.g_dirty = false;
    }
    if (wk.in_flight)
      throw std::runtime_error(
          "native engine invariant: dispatch while a round is in flight "
          "(worker " + std::to_string(wid) + ")");
    wk.round_serial += 1;
    wk.in_flight = true;
  }

  void dispatch_wave(const std::vector<int>& ready, double t_now) {
    cur_wst = wstreams[wave_rr % NWSTREAM];
    wave_rr += 1;
    HIP_CHECK(hipStreamWaitEvent(cur_wst, update_ev, 0));
    if (commit_slots_dev) {
      // SAGA: one commit + staging-reset kernel for the whole wave,
      // stream-ordered before the wave's gradient kernel
      CsrCommitCmd cc;
      cc.n = 0;
      cc.bper = 16;
      for (int wid : ready) {
        cc.wid[cc.n] = wid;
        cc.do_commit[cc.n] = ws[wid].pending_commit ? 1 : 0;
        ws[wid].pending_commit = false;
        cc.n += 1;
      }
      if (cc.n > 0) {
        launch_saga_commit_wave(commit_slots_dev, &cc, cur_wst);
        HIP_CHECK(hipGetLastError());
      }
    }
    GradWaveCmd cmd;
    cmd.n = 0;
    cmd.bper = wave_bper;
    cmd.interleave = wave_interleave;
    for (int wid : ready) {
      dispatch_book(wid, t_now);
      WorkerBuf& wk = ws[wid];
      cmd.wid[cmd.n] = wid;
      cmd.round_k[cmd.n] = (unsigned int)(wk.k_submit + 1);
      cmd.done_val[cmd.n] = wk.round_serial;
      cmd.n += 1;
    }
    if (cmd.n > 0) {
      if (slots_dev) {
        if (wave_spill) {
          // spill refresh, stream-ordered: recompute the round's Philox
          // row sets, then gather those master entries into the staging
          // tables the gradient reads
          launch_scan_rows_wave(slots_dev, &cmd, cfg.seed, cfg.rate,
                                cur_wst);
          launch_alpha_gather_wave(slots_dev, &cmd, cur_wst);
        }
        launch_grad_dense_wave(slots_dev, &cmd, wave_max_rows, cfg.d,
                               cfg.seed, cfg.rate, cfg.objective,
                               ws[0].x_is_bf16, cfg.algo == 1 ? 1 : 0,
                               cur_wst);
      } else {
        launch_grad_csr_wave(csr_slots_dev, &cmd, cfg.seed, cfg.rate,
                             cfg.objective, ws[0].x_is_bf16,
                             cfg.algo == 1 ? 1 : 0, cur_wst);
      }
      HIP_CHECK(hipGetLastError());
    }
  }

  // Pop every pending worker that may dispatch now: stragglers move to the
  // delayed queue with a due time, the rest land in ``ready`` (dispatched
  // after the batch flush writes their weight snapshots).
  void collect_ready(double t_now, std::vector<int>& ready) {
    if (pendingq.empty()) return;
    if (available() < gate()) return;
    // delay calibration activation (reference :247-252)
    if (!delay_flag && k > cfg.calib_window) {
      if (cul_count > 0) avg_delay_ms = cul_time_ms / cul_count;
      delay_flag = true;
    }
    const size_t qn = pendingq.size();
    for (size_t i = 0; i < qn; ++i) {
      const int wid = pendingq.front();
      pendingq.pop_front();
      const double dly = delay_ms_for(wid, k) / 1000.0;
      if (dly > 0) {
        ws[wid].busy = true;  // occupied while "straggling"
        delayed.emplace_back(t_now + dly, wid);
      } else {
        ready.push_back(wid);
      }
    }
  }

  // Completion bookkeeping only (no launches): arrival clock, staleness,
  // EWMA, tau filter, requeue. The caller batches accepted gradients into
  // one multi_update kernel per poll sweep.
  bool book_completion(int wid, double t_now) {
    WorkerBuf& wk = ws[wid];
    wk.busy = false;
    wk.in_flight = false;
    wk.finish_t = t_now;
    wk.tasks += 1;
    const double rt = t_now - wk.submit_t;
    const int staleness = clock - wk.ts;  // arrival-clock staleness
    clock += 1;
    max_staleness_seen = std::max<long>(max_staleness_seen, staleness);
    // ASGD: completion-clock staleness <= taw (SparkASGDThread.scala:172).
    // ASAGA: k - ts <= taw with ts = the arrival clock at submit
    // (SparkASAGAThread.scala:191, raw-ts packing RDD.scala:1333) — matches
    // server.py accepts() and csrc/server_dist.cpp, NOT the applied-update
    // counter at submit (k_submit), which would diverge once rejections occur.
    const bool accept = (cfg.algo == 1) ? (k - wk.ts) <= cfg.taw
                                        : staleness <= cfg.taw;
    if (accept) {
      if (k < cfg.calib_window) {
        cul_time_ms += rt * 1000.0;
        cul_count += 1;
      }
    } else {
      rejected += 1;
      wk.pending_commit = false;
      wk.g_dirty = true;  // dispatch re-zeroes before the next round
    }
    pendingq.push_back(wid);
    return accept;
  }

  // -- batched update assembly (one kernel per poll sweep; elementwise-
  //    sequential application inside the kernel makes the result identical
  //    to launching the per-round update kernels back to back) ------------
  MultiUpdateArgs mu{};

  void flush_batch(const std::vector<int>* snap_targets) {
    size_t si = 0;
    const size_t total = snap_targets ? snap_targets->size() : 0;
    bool launched = false;
    while (true) {
      mu.m = 0;
      while (si < total && mu.m < MU_MAX)
        mu.sw[mu.m++] = (*snap_targets)[si++];
      if (mu.n == 0 && mu.m == 0) break;
      mu.algo = cfg.algo;
      launch_multi_update((float*)w, (float* const*)g_tab_dev,
                          (float* const*)wbuf_tab_dev, (float*)alpha_bar,
                          (float)cfg.gamma, (float)inv_batch, (float)inv_N,
                          cfg.d, &mu, sstream);
      HIP_CHECK(hipGetLastError());
      launched = true;
      mu.n = 0;
      if (si >= total) break;
    }
    mu.m = 0;
    if (launched) HIP_CHECK(hipEventRecord(update_ev, sstream));
  }

  // Append one accepted gradient; handles snapshot/mark boundaries so the
  // exactly-K-steps bench contract and the printer_freq optVars cadence
  // keep per-update precision despite batching.
  void append_accepted(int wid, double t_now) {
    WorkerBuf& wk = ws[wid];
    const bool snap_now =
        cfg.snap_every > 0 && k % cfg.snap_every == 0 &&
        (long)snap_ms.size() < cfg.snap_cap;
    if (cfg.algo == 0) {
      const double gamma_k = cfg.gamma / std::sqrt((double)(k / cfg.P + 1));
      mu.scale[mu.n] = (float)(gamma_k * inv_batch);
    }
    mu.gw[mu.n] = wid;
    mu.n += 1;
    if (cfg.algo == 1) wk.pending_commit = true;
    k += 1;
    applied += 1;
    if (snap_now || k == cfg.mark_lo || k == cfg.mark_hi || mu.n == MU_MAX)
      flush_batch(nullptr);
    if (snap_now) {
      // optVars snapshot (reference SparkASGDThread.scala:195-198): after
      // the update that landed at the pre-increment printer_freq multiple
      HIP_CHECK(hipMemcpyAsync(
          (void*)(cfg.snap_ring + (uintptr_t)snap_ms.size() *
                                      (size_t)cfg.d * 4),
          (const void*)w, (size_t)cfg.d * 4, hipMemcpyDeviceToDevice,
          sstream));
      snap_ms.push_back((t_now - run_t0) * 1000.0);
    }
    if (k == cfg.mark_lo) {
      HIP_CHECK(hipStreamSynchronize(sstream));
      mark_lo_t = now_s();
    }
    if (k == cfg.mark_hi) {
      HIP_CHECK(hipStreamSynchronize(sstream));
      mark_hi_t = now_s();
    }
  }

  struct Result {
    long k = 0, applied = 0, rejected = 0, max_staleness = -1;
    double elapsed_ms = 0, mark_lo_t = 0, mark_hi_t = 0, avg_delay = 0;
    std::vector<double> waits;
    std::vector<double> snap_ms;
  };

  Result run() {
    HIP_CHECK(hipStreamCreateWithFlags(&sstream, hipStreamNonBlocking));
    HIP_CHECK(hipEventCreateWithFlags(&update_ev, hipEventDisableTiming));
    HIP_CHECK(hipEventRecord(update_ev, sstream));
    // completion flags: one pinned-host 128-B line per worker (the grad
    // kernel writes it with a system-scope release; default hipHostMalloc
    // memory is fine-grained coherent on MI355X) + one padded device
    // arrival counter per worker
    HIP_CHECK(hipHostMalloc((void**)&flags_host,
                            (size_t)cfg.P * 16 * sizeof(unsigned long long),
                            hipHostMallocDefault));
    memset((void*)flags_host, 0,
           (size_t)cfg.P * 16 * sizeof(unsigned long long));
    HIP_CHECK(hipMalloc((void**)&arr_dev,
                        (size_t)cfg.P * 16 * sizeof(unsigned long long)));
    HIP_CHECK(hipMemset(arr_dev, 0,
                        (size_t)cfg.P * 16 * sizeof(unsigned long long)));
    for (int i = 0; i < cfg.P; ++i) {
      ws[i].done_flag = flags_host + (size_t)i * 16;
      ws[i].done_arr = arr_dev + (size_t)i * 16;
      ws[i].round_serial = 0;
    }
    for (auto& wk : ws) {
      HIP_CHECK(hipStreamCreateWithFlags(&wk.stream, hipStreamNonBlocking));
    }
    // wave dispatch tables. Dense: ASGD pipe-kernel shapes. CSR: ASGD and
    // SAGA with device-resident history (spill mode keeps per-worker
    // launches — its per-round scan/gather chain doesn't batch).
    const bool no_wave = std::getenv("ASYNCAMD_NO_WAVE") != nullptr;
    const bool dense_wave =
        !ws[0].sparse && cfg.d % 4 == 0 && cfg.d <= 2048;
    const bool csr_wave = ws[0].sparse;
    wave_ok = (dense_wave || csr_wave) && cfg.P <= GRAD_WAVE_MAXP &&
              !no_wave;
    wave_spill = cfg.algo == 1 && ws[0].alpha_host != 0;
    if (wave_ok) {
      wave_max_rows = 0;
      for (int i = 0; i < cfg.P; ++i)
        wave_max_rows = std::max(wave_max_rows, ws[i].n_rows);
      if (dense_wave) {
        std::vector<GradWaveSlot> hs(cfg.P);
        for (int i = 0; i < cfg.P; ++i) {
          hs[i].X = (const void*)ws[i].X;
          hs[i].y = (const float*)ws[i].y;
          hs[i].wbuf = (const float*)ws[i].wbuf;
          hs[i].g = (float*)ws[i].g;
          hs[i].n_out = (int*)ws[i].ctr;
          hs[i].n_rows = ws[i].n_rows;
          hs[i].row_start = ws[i].row_start;
          hs[i].done_flag = (unsigned long long*)ws[i].done_flag;
          hs[i].done_arr = ws[i].done_arr;
          hs[i].alpha = (float*)ws[i].alpha;
          hs[i].idx_out = (int*)ws[i].idx_out;
          hs[i].e_out = (float*)ws[i].e_out;
          hs[i].pos_ctr = (int*)(ws[i].ctr + 4);
          hs[i].alpha_host = (const float*)ws[i].alpha_host;
          hs[i].srows = (int*)ws[i].srows;
          hs[i].sylist = (float*)ws[i].sylist;
          hs[i].scnt = (int*)ws[i].scnt;
        }
        HIP_CHECK(hipMalloc((void**)&slots_dev,
                            sizeof(GradWaveSlot) * cfg.P));
        HIP_CHECK(hipMemcpy(slots_dev, hs.data(),
                            sizeof(GradWaveSlot) * cfg.P,
                            hipMemcpyHostToDevice));
      } else {
        std::vector<CsrWaveSlot> hs(cfg.P);
        for (int i = 0; i < cfg.P; ++i) {
          hs[i].indptr = (const int*)ws[i].indptr;
          hs[i].indices = (const int*)ws[i].indices;
          hs[i].values = (const void*)ws[i].values;
          hs[i].y = (const float*)ws[i].y;
          hs[i].wbuf = (const float*)ws[i].wbuf;
          hs[i].alpha = (float*)ws[i].alpha;
          hs[i].g = (float*)ws[i].g;
          hs[i].n_out = (int*)ws[i].ctr;
          hs[i].idx_out = (int*)ws[i].idx_out;
          hs[i].e_out = (float*)ws[i].e_out;
          hs[i].pos_ctr = (int*)(ws[i].ctr + 4);
          hs[i].n_rows = ws[i].n_rows;
          hs[i].row_start = ws[i].row_start;
          hs[i].done_flag = (unsigned long long*)ws[i].done_flag;
          hs[i].done_arr = ws[i].done_arr;
        }
        HIP_CHECK(hipMalloc((void**)&csr_slots_dev,
                            sizeof(CsrWaveSlot) * cfg.P));
        HIP_CHECK(hipMemcpy(csr_slots_dev, hs.data(),
                            sizeof(CsrWaveSlot) * cfg.P,
                            hipMemcpyHostToDevice));
      }
      if (cfg.algo == 1) {  // unified commit view (dense + CSR SAGA)
        std::vector<CommitSlot> cs(cfg.P);
        for (int i = 0; i < cfg.P; ++i) {
          cs[i].dst = (float*)(ws[i].alpha_host ? ws[i].alpha_host
                                                : ws[i].alpha);
          cs[i].idx = (int*)ws[i].idx_out;
          cs[i].e = (float*)ws[i].e_out;
          cs[i].pos_ctr = (int*)(ws[i].ctr + 4);
          cs[i].scnt = ws[i].alpha_host ? (int*)ws[i].scnt : nullptr;
          cs[i].n_out = (int*)ws[i].ctr;
          cs[i].arr = ws[i].done_arr;
        }
        HIP_CHECK(hipMalloc((void**)&commit_slots_dev,
                            sizeof(CommitSlot) * cfg.P));
        HIP_CHECK(hipMemcpy(commit_slots_dev, cs.data(),
                            sizeof(CommitSlot) * cfg.P,
                            hipMemcpyHostToDevice));
      }
      for (int i = 0; i < NWSTREAM; ++i)
        HIP_CHECK(hipStreamCreateWithFlags(&wstreams[i],
                                           hipStreamNonBlocking));
      wave_bper = query_grad_grid(wave_max_rows);
      if (!std::getenv("ASYNCAMD_GRAD_GRID")) {
        // measured (flagship, P=32): with interleaved full-P waves the
        // chip is best fed by ~512-768 TOTAL blocks, each carrying many
        // sampled rows (deep pipelines) — per-worker grids sized as if
        // launched alone left 5 rows/block and 114k updates/s vs 186k
        // at 16-24 blocks/worker (same-box sweep, ASYNCAMD_GRAD_GRID)
        wave_bper = std::min(wave_bper, std::max(8, 512 / cfg.P));
      }
      // interleaved block->slot mapping measured +21% on the flagship
      // (workers progress together => completions bunch => bigger update
      // batches and fuller next waves)
      const char* wi = std::getenv("ASYNCAMD_WAVE_INTERLEAVE");
      wave_interleave = wi ? std::atoi(wi) : 1;
    }
    inv_batch = (double)cfg.P / (cfg.rate * (double)cfg.N);
    inv_N = 1.0 / (double)cfg.N;
    {  // device pointer tables for the batched update kernel
      std::vector<float*> hg(cfg.P), hw(cfg.P);
      for (int i = 0; i < cfg.P; ++i) {
        hg[i] = (float*)ws[i].g;
        hw[i] = (float*)ws[i].wbuf;
      }
      HIP_CHECK(hipMalloc(&g_tab_dev, cfg.P * sizeof(float*)));
      HIP_CHECK(hipMalloc(&wbuf_tab_dev, cfg.P * sizeof(float*)));
      HIP_CHECK(hipMemcpy(g_tab_dev, hg.data(), cfg.P * sizeof(float*),
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMemcpy(wbuf_tab_dev, hw.data(), cfg.P * sizeof(float*),
                          hipMemcpyHostToDevice));
    }
    init_stragglers();
    for (int i = 0; i < cfg.P; ++i) pendingq.push_back(i);
    const double t0 = now_s();
    run_t0 = t0;
    // first dispatch ignores the gate (reference k==0 path)
    {
      const size_t qn = pendingq.size();
      for (size_t i = 0; i < qn; ++i) {
        const int wid = pendingq.front();
        pendingq.pop_front();
        dispatch(wid, t0);
      }
    }
    double last_progress = t0;
    std::vector<int> ready;
    ready.reserve(cfg.P);
    while (k < cfg.iters) {
      const double t_now = now_s();
      if (t_now - t0 > cfg.max_wall_s) break;
      // stall watchdog: a wedged loop (lost completion, bad event, driver
      // hang) must fail loudly in seconds, not spin GPU-idle to max_wall_s
      // (the round-1 failure mode). Delayed dispatches push the deadline.
      double stall_deadline = last_progress + cfg.stall_s;
      for (auto& dw : delayed)
        stall_deadline = std::max(stall_deadline, dw.first + cfg.stall_s);
      if (t_now > stall_deadline) {
        int busy_n = 0;
        for (auto& wk : ws) busy_n += wk.busy ? 1 : 0;
        throw std::runtime_error(
            "native engine stalled: no completion for " +
            std::to_string(cfg.stall_s) + " s at k=" + std::to_string(k) +
            " (busy=" + std::to_string(busy_n) +
            ", delayed=" + std::to_string(delayed.size()) +
            ", pending=" + std::to_string(pendingq.size()) + ")");
      }
      // release due delayed dispatches (due times are NOT monotone across
      // the deque — straggler draws differ per worker — so scan all)
      for (size_t di = 0; di < delayed.size();) {
        if (delayed[di].first <= t_now) {
          const int wid = delayed[di].second;
          delayed.erase(delayed.begin() + (long)di);
          dispatch(wid, t_now);
        } else {
          ++di;
        }
      }
      // poll completions; accepted gradients accumulate into one batched
      // update kernel per sweep (append_accepted flushes early at snapshot
      // and bench-mark boundaries to keep per-update precision)
      bool any = false;
      for (int i = 0; i < cfg.P && k < cfg.iters; ++i) {
        WorkerBuf& wk = ws[i];
        // NB round-1 regression: a `submit_t > finish_t` guard here was
        // permanently false after dispatch() seeded finish_t = t_now on the
        // first round (BENCH_r01 30-min 0%-GPU hang). !busy alone gates
        // polling; busy is only set between dispatch and completion.
        if (!wk.busy || !wk.in_flight) continue;
        // event-free completion: the grad kernel's last block published
        // round_serial to this pinned line (system-scope release store);
        // a plain volatile read replaces hipEventQuery
        if (*wk.done_flag == wk.round_serial) {
          const double tc = now_s();
          if (book_completion(i, tc)) append_accepted(i, tc);
          any = true;
        }
      }
      if (any) {
        last_progress = t_now;
        ready.clear();
        collect_ready(now_s(), ready);
        // one kernel: apply the sweep's remaining accepted gradients AND
        // write the post-batch w into the redispatching workers' snapshot
        // buffers (replaces a 3 KB hipMemcpyAsync per dispatch)
        flush_batch(&ready);
        const double td = now_s();
        if (wave_ok)
          dispatch_wave(ready, td);
        else
          for (int wid : ready) dispatch_impl(wid, td, false);
      }
    }
    HIP_CHECK(hipStreamSynchronize(sstream));
    const double t1 = now_s();
    // drain in-flight rounds so buffers are quiescent before Python
    // resumes — INCLUDING the wave streams: the final sweep may have
    // dispatched a wave whose kernels are still writing g/wbuf/alpha and
    // publishing to the pinned flags this teardown is about to free
    for (auto& wk : ws) HIP_CHECK(hipStreamSynchronize(wk.stream));
    for (int i = 0; i < NWSTREAM; ++i)
      if (wstreams[i]) HIP_CHECK(hipStreamSynchronize(wstreams[i]));
    Result out;
    out.k = k;
    out.elapsed_ms = (t1 - t0) * 1000.0;
    out.applied = applied;
    out.rejected = rejected;
    out.max_staleness = max_staleness_seen;
    out.mark_lo_t = mark_lo_t;
    out.mark_hi_t = mark_hi_t;
    out.avg_delay = avg_delay_ms;
    out.snap_ms = snap_ms;
    for (auto& wk : ws) out.waits.push_back(wk.waiting_ms);
    for (auto& wk : ws) {
      HIP_CHECK(hipStreamDestroy(wk.stream));
      wk.done_flag = nullptr;
      wk.done_arr = nullptr;
    }
    HIP_CHECK(hipHostFree((void*)flags_host));
    flags_host = nullptr;
    HIP_CHECK(hipFree(arr_dev));
    arr_dev = nullptr;
    if (slots_dev) {
      HIP_CHECK(hipFree(slots_dev));
      slots_dev = nullptr;
    }
    if (csr_slots_dev) {
      HIP_CHECK(hipFree(csr_slots_dev));
      csr_slots_dev = nullptr;
    }
    if (commit_slots_dev) {
      HIP_CHECK(hipFree(commit_slots_dev));
      commit_slots_dev = nullptr;
    }
    for (int i = 0; i < NWSTREAM; ++i)
      if (wstreams[i]) {
        HIP_CHECK(hipStreamDestroy(wstreams[i]));
        wstreams[i] = nullptr;
      }
    HIP_CHECK(hipEventDestroy(update_ev));
    HIP_CHECK(hipStreamDestroy(sstream));
    HIP_CHECK(hipFree(g_tab_dev));
    HIP_CHECK(hipFree(wbuf_tab_dev));
    g_tab_dev = wbuf_tab_dev = nullptr;
    return out;
  }
};

}  // namespace

void register_native_engine(py::module_& m) {
  m.def(
      "native_local_run",
      [](py::dict c, py::list workers, uintptr_t w, uintptr_t alpha_bar) {
        NativeEngine eng;
        EngineCfg& cfg = eng.cfg;
        cfg.N = py::cast<long>(c["N"]);
        cfg.d = py::cast<int>(c["d"]);
        cfg.P = py::cast<int>(c["P"]);
        cfg.iters = py::cast<long>(c["iters"]);
        cfg.gamma = py::cast<double>(c["gamma"]);
        cfg.rate = py::cast<double>(c["rate"]);
        cfg.bucket_ratio = py::cast<double>(c["bucket_ratio"]);
        cfg.taw = py::cast<long>(c["taw"]);
        cfg.seed = py::cast<uint64_t>(c["seed"]);
        cfg.algo = py::cast<int>(c["algo"]);
        cfg.objective = py::cast<int>(c["objective"]);
        cfg.coeff = py::cast<double>(c["coeff"]);
        cfg.calib_window = py::cast<long>(c["calib_window"]);
        cfg.mark_lo = py::cast<long>(c["mark_lo"]);
        cfg.mark_hi = py::cast<long>(c["mark_hi"]);
        cfg.max_wall_s = py::cast<double>(c["max_wall_s"]);
        if (c.contains("stall_s"))
          cfg.stall_s = py::cast<double>(c["stall_s"]);
        cfg.snap_every = py::cast<long>(c["snap_every"]);
        cfg.snap_ring = py::cast<uintptr_t>(c["snap_ring"]);
        cfg.snap_cap = py::cast<long>(c["snap_cap"]);
        for (auto item : workers) {
          py::dict wd = py::cast<py::dict>(item);
          WorkerBuf wk;
          wk.sparse = py::cast<bool>(wd["sparse"]);
          if (wk.sparse) {
            wk.indptr = py::cast<uintptr_t>(wd["indptr"]);
            wk.indices = py::cast<uintptr_t>(wd["indices"]);
            wk.values = py::cast<uintptr_t>(wd["values"]);
          } else {
            wk.X = py::cast<uintptr_t>(wd["X"]);
          }
          wk.y = py::cast<uintptr_t>(wd["y"]);
          wk.wbuf = py::cast<uintptr_t>(wd["wbuf"]);
          wk.g = py::cast<uintptr_t>(wd["g"]);
          wk.ctr = py::cast<uintptr_t>(wd["ctr"]);
          wk.n_rows = py::cast<long>(wd["n_rows"]);
          wk.row_start = py::cast<long>(wd["row_start"]);
          wk.x_is_bf16 = py::cast<int>(wd["x_is_bf16"]);
          if (eng.cfg.algo == 1) {
            wk.alpha = py::cast<uintptr_t>(wd["alpha"]);
            wk.idx_out = py::cast<uintptr_t>(wd["idx_out"]);
            wk.e_out = py::cast<uintptr_t>(wd["e_out"]);
            wk.saga_cap = py::cast<int>(wd["saga_cap"]);
            if (wd.contains("alpha_host")) {
              wk.alpha_host = py::cast<uintptr_t>(wd["alpha_host"]);
              wk.srows = py::cast<uintptr_t>(wd["srows"]);
              wk.sylist = py::cast<uintptr_t>(wd["sylist"]);
              wk.scnt = py::cast<uintptr_t>(wd["scnt"]);
            }
          }
          eng.ws.push_back(wk);
        }
        if ((int)eng.ws.size() != cfg.P)
          throw std::runtime_error("native engine: P != len(workers)");
        eng.w = w;
        eng.alpha_bar = alpha_bar;
        NativeEngine::Result r;
        {
          py::gil_scoped_release rel;  // the event loop never touches Python
          r = eng.run();
        }
        py::dict out;
        out["k"] = r.k;
        out["elapsed_ms"] = r.elapsed_ms;
        out["applied"] = r.applied;
        out["rejected"] = r.rejected;
        out["max_staleness"] = r.max_staleness;
        out["mark_lo_t"] = r.mark_lo_t;
        out["mark_hi_t"] = r.mark_hi_t;
        out["avg_delay_ms"] = r.avg_delay;
        out["waiting_ms"] = r.waits;
        out["snap_ms"] = r.snap_ms;
        return out;
      });
}
