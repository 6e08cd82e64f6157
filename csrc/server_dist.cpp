// Native multi-GPU parameter server: the rank-0 control plane of the dist
// engine in C++ (ROADMAP item 1).
//
// The Python dist engine (engine/dist.py) spends ~100-250 us of GIL-bound
// Python per update (proxy threads, pack/unpack, server bookkeeping); this
// server runs the same protocol with zero Python on the hot path:
//
//   * one C++ channel thread per REMOTE worker drives the pair
//     ProcessGroup (send dispatch -> recv result) — under the "nccl"
//     backend that is RCCL point-to-point over xGMI; under gloo it runs on
//     CPU, which is how the CPU test tier exercises THIS exact loop.
//   * rank-0-local workers call in through local_next_dispatch /
//     local_deliver (GIL released while blocked), so the Python worker
//     objects (HIP gradient kernels on their own streams) plug in
//     unchanged.
//   * tau filter, quorum gate, straggler model, update rules and
//     bookkeeping mirror engine/local.py::AsyncEngine + engine/server.py
//     exactly (reference semantics: SparkASGDThread.scala:153-345,
//     SparkASAGAThread.scala:191,217-220, RDD.scala:1144-1165).
//
// WIRE-COMPATIBLE with engine/dist.py's remote_worker_loop: the packed
// [d + 8] float layout of engine/messages.py (H_TS..H_SNAP), tag 0, peer
// group-rank 1. Worker ranks keep running the validated Python loop.
//
// Updates use aten tensor ops (w.add_ etc.) — identical numerics to
// ops.torch_ref on CPU and cuBLAS-free elementwise HIP kernels on GPU; the
// fused-update HIP kernels remain available for the single-GPU native
// engine. The checkpoint snap sideband (engine/dist.py protocol: snap=1 ->
// worker sends its SAGA history table) is wired: request_alpha_snapshot /
// wait_alpha gather remote tables mid-run; resume pushes tables back from
// Python BEFORE start() (no channel thread is running yet, so plain
// dist.send on the pair group is race-free).

#include <torch/extension.h>
#ifdef __HIP_PLATFORM_AMD__
#include <c10/hip/HIPStream.h>
extern "C" void launch_dist_sgd_update(float*, const float*, float, float,
                                       int, hipStream_t);
extern "C" void launch_dist_saga_update(float*, const float*, float*, float,
                                        float, float, int, hipStream_t);
namespace at { namespace cuda {
using c10::hip::getCurrentHIPStream;
} }
#endif
#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>

#include <atomic>
#include <chrono>
#include <cmath>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <map>
#include <mutex>
#include <set>
#include <thread>
#include <vector>

namespace {

constexpr int HDR = 8;
enum { H_TS = 0, H_K, H_ACCEPT, H_STOP, H_DELAY, H_NROWS, H_ELAPSED, H_SNAP };

// ProcessGroup op ENQUEUES are serialized exactly like engine/dist.py's
// _PG_LOCK (torch's backend bindings are not documented thread-safe for
// concurrent enqueue); Work::wait runs OUTSIDE the lock so channels still
// progress concurrently.
std::mutex g_pg_mu;

inline c10::intrusive_ptr<c10d::Work> pg_send(
    const c10::intrusive_ptr<c10d::ProcessGroup>& pg,
    std::vector<at::Tensor>& v, int dst, int tag) {
  std::lock_guard<std::mutex> lk(g_pg_mu);
  return pg->send(v, dst, tag);
}

inline c10::intrusive_ptr<c10d::Work> pg_recv(
    const c10::intrusive_ptr<c10d::ProcessGroup>& pg,
    std::vector<at::Tensor>& v, int src, int tag) {
  std::lock_guard<std::mutex> lk(g_pg_mu);
  return pg->recv(v, src, tag);
}

inline double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// Host Philox x0 (bit-identical to csrc/philox.h / utils/philox.py) for the
// reproducible straggler draws: uniform01 counters (0, 0, round, stream).
inline uint32_t philox_host_x0(uint64_t seed, uint32_t c0, uint32_t c1,
                               uint32_t c2, uint32_t c3) {
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
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
  return c0;
}

inline double uniform01_host(uint64_t seed, uint32_t round_k,
                             uint32_t stream) {
  return philox_host_x0(seed, 0u, 0u, round_k, stream) / 4294967296.0;
}

struct DSCfg {
  int64_t d = 0, P = 0, M = 0;  // dims, logical workers, rank-0 locals
  int64_t N = 0, num_iter = 0, printer_freq = 100;
  double gamma = 0.01, batch_rate = 0.1;
  int64_t taw = 1 << 30;
  int64_t gate = 0;
  double coeff = 0.0;
  uint64_t seed = 42;
  int64_t calib_window = 0;
  bool asaga = false;
  bool snapshot_weights = false;
  int64_t k0 = 0, clock0 = 0;  // resume-from-checkpoint initial state
  double bucket_ratio = 0.93;   // for the alive-scaled quorum gate
  bool trace = false;           // record dispatch/accept/reject events
  double worker_timeout_s = 0;  // failure detection (0 = off), matching
                                // engine/local.py::_reap_dead_workers
  double par_recs() const { return batch_rate * (double)N / (double)P; }
};

struct DispatchMsg {
  at::Tensor w;  // shared snapshot (all dispatches of a round share one)
  int64_t ts = 0, k = 0;
  bool accept = true, stop = false;
  double delay_s = 0.0;
  int snap = 0;  // 1 = fetch the peer's SAGA history (checkpoint sideband)
};

struct Slot {
  std::mutex m;
  std::condition_variable cv;
  std::deque<DispatchMsg> q;  // regular round + snap ops interleave FIFO
};

class DistServer {
 public:
  DistServer(DSCfg cfg, at::Tensor w0,
             std::vector<c10::intrusive_ptr<c10d::ProcessGroup>> pair_pgs,
             std::vector<int64_t> mark_at,
             std::vector<int64_t> alpha_rows = {})
      : cfg_(cfg), w_(w0), pgs_(std::move(pair_pgs)) {
    TORCH_CHECK(w_.dtype() == at::kFloat && w_.numel() == cfg_.d);
    TORCH_CHECK((int64_t)pgs_.size() == cfg_.P - cfg_.M,
                "one pair ProcessGroup per remote worker, wid order");
    k_ = cfg_.k0;
    clock_ = cfg_.clock0;
    alpha_rows_ = alpha_rows;
    alpha_rows_.resize(cfg_.P - cfg_.M, 0);
    alpha_snap_.resize(cfg_.P - cfg_.M);
    alpha_ready_.assign(cfg_.P - cfg_.M, 0);
    if (cfg_.asaga)
      alpha_bar_ = at::zeros({cfg_.d}, w_.options());
    avail_.assign(cfg_.P, 1);
    dead_.assign(cfg_.P, 0);
    last_accept_.assign(cfg_.P, 1);
    submit_t_.assign(cfg_.P, 0.0);
    finish_t_.assign(cfg_.P, 0.0);
    waiting_ms_.assign(cfg_.P, 0);
    for (int64_t i = 0; i < cfg_.P; ++i)
      slots_.emplace_back(new Slot());
    for (auto m : mark_at) mark_at_.insert(m);
    init_stragglers();
  }

  // ---- lifecycle ---------------------------------------------------------
  void start() {
    t0_ = now_s();
    if (cfg_.snapshot_weights)
      opt_ms_.push_back(0), opt_w_.push_back(w_.detach().cpu().clone());
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (k_ >= cfg_.num_iter) {  // resumed with the budget already spent
        finish_locked();
      } else {
        for (int64_t wid = 0; wid < cfg_.P; ++wid)
          pending_.push_back((int)wid);
        try_dispatch(/*first=*/true);
      }
    }
    for (int64_t wid = cfg_.M; wid < cfg_.P; ++wid)
      threads_.emplace_back(&DistServer::channel_loop, this, (int)wid);
  }

  bool wait_done(double timeout_s) {
    const double deadline = now_s() + timeout_s;
    std::unique_lock<std::mutex> lk(mu_);
    while (!done_ && now_s() < deadline) {
      done_cv_.wait_for(lk, std::chrono::milliseconds(100),
                        [&] { return done_; });
      if (!done_ && cfg_.worker_timeout_s > 0) reap_dead_locked();
    }
    if (!done_) finish_locked();  // wall-clock cap: stop everything
    return done_;
  }

  void join() {
    // a dead peer's channel thread can be blocked in recv forever — wait
    // up to 10 s for clean exits, then detach the stragglers (they only
    // wake, if ever, while this object is still alive in the engine)
    {
      std::unique_lock<std::mutex> lk(mu_);
      threads_exit_cv_.wait_for(lk, std::chrono::seconds(10), [&] {
        return threads_exited_ >= (int64_t)threads_.size();
      });
    }
    for (auto& t : threads_) {
      if (!t.joinable()) continue;
      if (threads_exited_ >= (int64_t)threads_.size())
        t.join();
      else
        t.detach();
    }
    threads_.clear();
  }

  std::string channel_error() const { return channel_error_; }

  // (ts_seconds_monotonic, wid, kind 0=dispatch/1=accept/2=reject, k,
  //  staleness) — merged into the Perfetto log by the Python adapter
  std::vector<std::tuple<double, int64_t, int64_t, int64_t, int64_t>>
  trace_events() {
    std::lock_guard<std::mutex> lk(mu_);
    std::vector<std::tuple<double, int64_t, int64_t, int64_t, int64_t>> out;
    out.reserve(trace_ev_.size());
    for (auto& e : trace_ev_)
      out.emplace_back(e.ts, e.wid, e.kind, e.k, e.staleness);
    return out;
  }

  int64_t dead_workers() const {
    int64_t n = 0;
    for (auto d : dead_) n += d;
    return n;
  }

  // ---- rank-0-local worker API (called from Python worker threads) -------
  // returns (w, ts, k_submit, accept_prev, delay_s, stop)
  std::tuple<at::Tensor, int64_t, int64_t, bool, double, bool>
  local_next_dispatch(int64_t wid) {
    Slot& s = *slots_[wid];
    std::unique_lock<std::mutex> lk(s.m);
    s.cv.wait(lk, [&] { return !s.q.empty(); });
    DispatchMsg m = s.q.front();
    s.q.pop_front();
    return {m.w, m.ts, m.k, m.accept, m.delay_s, m.stop};
  }

  // ---- checkpoint sideband (SAGA history gather, engine/dist.py snap=1) --
  void request_alpha_snapshot(int64_t wid) {
    TORCH_CHECK(wid >= cfg_.M && wid < cfg_.P, "remote wids only");
    TORCH_CHECK(alpha_rows_[wid - cfg_.M] > 0, "no history table for wid");
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (done_) {
        // shutdown already queued a stop: the peer can no longer answer —
        // resolve as empty so the checkpoint thread never blocks
        alpha_snap_[wid - cfg_.M] = at::Tensor();
        alpha_ready_[wid - cfg_.M] = 1;
        alpha_cv_.notify_all();
        return;
      }
      alpha_ready_[wid - cfg_.M] = 0;
    }
    DispatchMsg m;
    m.snap = 1;
    fill_slot((int)wid, m);
  }

  bool wait_alpha(int64_t wid, double timeout_s) {
    std::unique_lock<std::mutex> lk(mu_);
    return alpha_cv_.wait_for(lk, std::chrono::duration<double>(timeout_s),
                              [&] { return alpha_ready_[wid - cfg_.M] != 0; });
  }

  at::Tensor get_alpha(int64_t wid) {
    std::lock_guard<std::mutex> lk(mu_);
    return alpha_snap_[wid - cfg_.M];
  }

  int64_t clock() const { return clock_; }
  at::Tensor alpha_bar() const { return alpha_bar_; }

  void local_deliver(int64_t wid, at::Tensor g, int64_t ts, int64_t k_submit,
                     double elapsed_ms) {
    deliver((int)wid, g, ts, k_submit, elapsed_ms);
  }

  // ---- results -----------------------------------------------------------
  int64_t k() const { return k_; }
  int64_t applied() const { return applied_; }
  int64_t rejected() const { return rejected_; }
  int64_t elapsed_ms() const {
    // frozen at finish (matches the Python engines, which stamp elapsed
    // when the run loop exits, not when results are collected)
    return done_ ? elapsed_final_ms_
                 : (int64_t)((now_s() - t0_) * 1000.0);
  }
  std::vector<int64_t> waiting_ms() const { return waiting_ms_; }
  std::map<int64_t, double> marks() const { return marks_; }
  std::vector<int64_t> opt_ms() const { return opt_ms_; }
  std::vector<at::Tensor> opt_w() const { return opt_w_; }
  at::Tensor weights() const { return w_; }
  int64_t max_staleness_seen() const { return max_staleness_; }
  bool delay_active() const { return delay_flag_; }
  double avg_delay_ms() const { return avg_delay_ms_; }

  // straggler-model probe for cross-checking against engine/delay.py
  static double delay_probe(int64_t P, double coeff, uint64_t seed,
                            double avg_ms, int64_t wid, int64_t round_k) {
    DSCfg c;
    c.P = P;
    c.coeff = coeff;
    c.seed = seed;
    DistServer* tmp = nullptr;
    (void)tmp;
    // replicate delay_ms_for with an ad-hoc straggler table
    std::vector<int> kind(P, 0);
    const int length = (int)std::lround(0.25 * (double)P);
    const int length_normal = (int)std::lround(0.8 * length);
    const int length_longtail = length - length_normal;
    for (int cc = 0; cc < length; ++cc) {
      const int idx = cc * 4;
      if (idx < P) kind[idx] = (cc < length_longtail) ? 2 : 1;
    }
    if (coeff == 0.0) return 0.0;
    if (coeff != -1.0)
      return (wid == 0 && coeff > 0) ? std::round(coeff * avg_ms) : 0.0;
    if (kind[wid] == 2)
      return std::round(
          (uniform01_host(seed, (uint32_t)round_k, (uint32_t)wid) * 7.5 +
           2.5) * avg_ms);
    if (kind[wid] == 1)
      return std::round(
          (uniform01_host(seed, (uint32_t)round_k, (uint32_t)wid) + 1.5) *
          avg_ms);
    return 0.0;
  }

 private:
  // ---- remote channel: one thread per remote worker ----------------------
  void channel_loop(int wid) {
    try {
      channel_loop_body(wid);
    } catch (const std::exception& e) {
      // transport failure (e.g. a dead peer closed the connection while we
      // were blocked in recv): the worker stays/becomes declared dead and
      // the channel retires — never let the exception reach
      // std::terminate on a detached thread
      std::lock_guard<std::mutex> lk(mu_);
      if (wid >= 0 && wid < (int)dead_.size()) dead_[wid] = 1;
      channel_error_ = e.what();
    } catch (...) {
      std::lock_guard<std::mutex> lk(mu_);
      channel_error_ = "unknown channel transport error";
    }
    {
      std::lock_guard<std::mutex> lk(mu_);
      threads_exited_ += 1;
    }
    threads_exit_cv_.notify_all();
  }

  void channel_loop_body(int wid) {
    auto pg = pgs_[wid - cfg_.M];
    at::Tensor buf = at::zeros({cfg_.d + HDR}, w_.options());
    at::Tensor hdr_host = at::zeros({HDR}, at::kFloat);
    Slot& s = *slots_[wid];
    std::vector<at::Tensor> v{buf};
    while (true) {
      DispatchMsg m;
      {
        std::unique_lock<std::mutex> lk(s.m);
        s.cv.wait(lk, [&] { return !s.q.empty(); });
        m = s.q.front();
        s.q.pop_front();
      }
      // pack (engine/messages.py layout; float32 header => integers
      // exact to 2^24, same bound the Python packer asserts)
      if (m.w.defined()) buf.narrow(0, 0, cfg_.d).copy_(m.w);
      hdr_host[H_TS] = (float)m.ts;
      hdr_host[H_K] = (float)m.k;
      hdr_host[H_ACCEPT] = m.accept ? 1.f : 0.f;
      hdr_host[H_STOP] = m.stop ? 1.f : 0.f;
      hdr_host[H_DELAY] = (float)m.delay_s;
      hdr_host[H_NROWS] = 0.f;
      hdr_host[H_ELAPSED] = 0.f;
      hdr_host[H_SNAP] = (float)m.snap;
      buf.narrow(0, cfg_.d, HDR).copy_(hdr_host);
      pg_send(pg, v, /*group-rank of peer*/ 1, /*tag*/ 0)->wait();
      if (m.stop) {
        // resolve any snaps queued behind the stop as empty. Lock order:
        // never hold s.m while taking mu_ (finish_locked holds mu_ and
        // fills slots -> s.m), so inspect the queue first, then mark.
        bool pending_snap = false;
        {
          std::lock_guard<std::mutex> lk(s.m);
          for (auto& rest : s.q)
            if (rest.snap == 1) pending_snap = true;
          s.q.clear();
        }
        if (pending_snap) {
          std::lock_guard<std::mutex> lk2(mu_);
          alpha_snap_[wid - cfg_.M] = at::Tensor();
          alpha_ready_[wid - cfg_.M] = 1;
        }
        alpha_cv_.notify_all();
        break;
      }
      if (m.snap == 1) {
        // checkpoint sideband: the peer replies with its history table
        at::Tensor ab = at::zeros({alpha_rows_[wid - cfg_.M]}, w_.options());
        std::vector<at::Tensor> av{ab};
        pg_recv(pg, av, 1, 0)->wait();
        {
          std::lock_guard<std::mutex> lk(mu_);
          alpha_snap_[wid - cfg_.M] = ab.cpu();
          alpha_ready_[wid - cfg_.M] = 1;
        }
        alpha_cv_.notify_all();
        continue;
      }
      pg_recv(pg, v, 1, 0)->wait();
      // header comes back to host (this is also the completion sync point)
      at::Tensor h = buf.narrow(0, cfg_.d, HDR).cpu();
      const float* hp = h.data_ptr<float>();
      deliver(wid, buf.narrow(0, 0, cfg_.d), (int64_t)hp[H_TS],
              (int64_t)hp[H_K], (double)hp[H_ELAPSED]);
    }
  }

  // ---- completion path (engine/local.py::_process_result +
  //      engine/server.py::on_completion/accepts/apply) --------------------
  void deliver(int wid, at::Tensor g, int64_t ts, int64_t k_submit,
               double elapsed_ms) {
    (void)k_submit;
    (void)elapsed_ms;
    std::lock_guard<std::mutex> lk(mu_);
    if (done_) return;  // late result after shutdown: dropped (as in Python)
    if (k_ >= cfg_.num_iter) {  // budget already reached (Python loop
      finish_locked();          // checks BEFORE processing — mirror that)
      return;
    }
    const double t_now = now_s();
    const int64_t staleness = clock_ - ts;  // arrival clock
    clock_ += 1;
    max_staleness_ = std::max(max_staleness_, staleness);
    avail_[wid] = 1;
    dead_[wid] = 0;  // a late result resurrects a declared-dead worker
    const bool accept =
        cfg_.asaga ? (k_ - ts) <= cfg_.taw : staleness <= cfg_.taw;
    if (cfg_.trace && trace_ev_.size() < kTraceCap)
      trace_ev_.push_back({t_now, (int32_t)wid, accept ? (int8_t)1
                                                       : (int8_t)2,
                           (int32_t)k_, (int32_t)staleness});
    if (accept) {
      finish_t_[wid] = t_now;
      // delay calibration sample (reference :177-186; Python records wall
      // round time on accepted results only)
      cul_time_ms_ += (t_now - submit_t_[wid]) * 1000.0;
      cul_count_ += 1;
      apply(g);
      last_accept_[wid] = 1;
      if (k_ % cfg_.printer_freq == 0 && cfg_.snapshot_weights) {
        opt_ms_.push_back((int64_t)((t_now - t0_) * 1000.0));
        opt_w_.push_back(w_.detach().cpu().clone());
      }
      k_ += 1;
      applied_ += 1;
      if (mark_at_.count(k_)) marks_[k_] = now_s();
    } else {
      last_accept_[wid] = 0;
      rejected_ += 1;
    }
    pending_.push_back(wid);
    if (k_ >= cfg_.num_iter) {
      finish_locked();
      return;
    }
    try_dispatch(false);
  }

  void apply(const at::Tensor& g) {
    at::Tensor gg = g;
    if (gg.device() != w_.device()) gg = gg.to(w_.device());
#ifdef __HIP_PLATFORM_AMD__
    if (w_.is_cuda()) {
      // one fused HIP launch on the current stream instead of 1-3 aten
      // dispatcher round-trips per update (same arithmetic as the
      // local-engine kernels; aten path below stays the CPU/gloo oracle)
      auto stream = at::cuda::getCurrentHIPStream().stream();
      if (cfg_.asaga)
        launch_dist_saga_update(
            w_.data_ptr<float>(), gg.data_ptr<float>(),
            alpha_bar_.data_ptr<float>(), (float)cfg_.gamma,
            (float)(1.0 / cfg_.par_recs()), (float)(1.0 / (double)cfg_.N),
            (int)cfg_.d, stream);
      else
        launch_dist_sgd_update(
            w_.data_ptr<float>(), gg.data_ptr<float>(),
            (float)(cfg_.gamma /
                    std::sqrt((double)(k_ / cfg_.P + 1))),
            (float)(1.0 / cfg_.par_recs()), (int)cfg_.d, stream);
      return;
    }
#endif
    if (cfg_.asaga) {
      // w -= gamma*(g/parRecs); w -= gamma*alphaBar; alphaBar += g/N
      // (SparkASAGAThread.scala:217-220)
      w_.add_(gg, -cfg_.gamma / cfg_.par_recs());
      w_.add_(alpha_bar_, -cfg_.gamma);
      alpha_bar_.add_(gg, 1.0 / (double)cfg_.N);
    } else {
      // gamma/sqrt(k/P + 1), Scala INT division (SparkASGDThread.scala:190)
      const double gamma_k =
          cfg_.gamma / std::sqrt((double)(k_ / cfg_.P + 1));
      w_.add_(gg, -gamma_k / cfg_.par_recs());
    }
  }

  // ---- failure detection (engine/local.py::_reap_dead_workers) -----------
  void reap_dead_locked() {
    const double now = now_s();
    bool changed = false;
    for (int64_t wid = 0; wid < cfg_.P; ++wid) {
      if (dead_[wid] || avail_[wid]) continue;
      if (submit_t_[wid] > 0 &&
          now - submit_t_[wid] > cfg_.worker_timeout_s) {
        dead_[wid] = 1;
        changed = true;
      }
    }
    if (changed) try_dispatch(false);  // a gate stall may now unblock
  }

  // ---- dispatch (engine/local.py::_dispatch_pending) ---------------------
  void try_dispatch(bool first) {
    if (pending_.empty()) return;
    int64_t avail_n = 0, ndead = 0;
    for (auto a : avail_) avail_n += a;
    for (auto d : dead_) ndead += d;
    const int64_t alive = cfg_.P - ndead;
    const int64_t gate = std::min(
        cfg_.gate,
        std::max((int64_t)1, (int64_t)(alive * cfg_.bucket_ratio)));
    const int64_t init_workers = first ? cfg_.P : avail_n;
    if (init_workers < gate) return;
    if (!delay_flag_ && k_ > cfg_.calib_window) {
      if (cul_count_ > 0) avg_delay_ms_ = cul_time_ms_ / (double)cul_count_;
      delay_flag_ = true;
    }
    at::Tensor w_snap = w_.detach().clone();
    const double t_now = now_s();
    const size_t qn = pending_.size();
    for (size_t i = 0; i < qn; ++i) {
      const int wid = pending_.front();
      pending_.pop_front();
      const double prev_fin = finish_t_[wid] == 0.0 ? t_now : finish_t_[wid];
      waiting_ms_[wid] += (int64_t)((t_now - prev_fin) * 1000.0);
      submit_t_[wid] = t_now;
      avail_[wid] = 0;
      if (cfg_.trace && trace_ev_.size() < kTraceCap)
        trace_ev_.push_back({t_now, (int32_t)wid, (int8_t)0, (int32_t)k_,
                             0});
      DispatchMsg m;
      m.w = w_snap;
      m.ts = clock_;
      m.k = k_;
      m.accept = last_accept_[wid] != 0;
      m.delay_s = delay_ms_for(wid, k_) / 1000.0;
      m.stop = false;
      fill_slot(wid, m);
    }
  }

  void fill_slot(int wid, const DispatchMsg& m) {
    Slot& s = *slots_[wid];
    std::lock_guard<std::mutex> lk(s.m);
    s.q.push_back(m);
    s.cv.notify_one();
  }

  void finish_locked() {
    if (done_) return;
    done_ = true;
    elapsed_final_ms_ = (int64_t)((now_s() - t0_) * 1000.0);
    DispatchMsg stop;
    stop.stop = true;
    stop.w = at::Tensor();
    for (int64_t wid = 0; wid < cfg_.P; ++wid) fill_slot((int)wid, stop);
    done_cv_.notify_all();
  }

  // ---- straggler model (reference SparkASGDThread.scala:124-141,287-312) -
  void init_stragglers() {
    straggler_kind_.assign(cfg_.P, 0);
    const int length = (int)std::lround(0.25 * (double)cfg_.P);
    const int length_normal = (int)std::lround(0.8 * length);
    const int length_longtail = length - length_normal;
    for (int c = 0; c < length; ++c) {
      const int idx = c * 4;
      if (idx < cfg_.P) straggler_kind_[idx] = (c < length_longtail) ? 2 : 1;
    }
  }

  double delay_ms_for(int wid, int64_t round_k) const {
    if (!delay_flag_ || cfg_.coeff == 0.0) return 0.0;
    if (cfg_.coeff != -1.0) {
      if (wid == 0 && cfg_.coeff > 0)
        return std::round(cfg_.coeff * avg_delay_ms_);
      return 0.0;
    }
    if (straggler_kind_[wid] == 2) {
      const double u =
          uniform01_host(cfg_.seed, (uint32_t)round_k, (uint32_t)wid);
      return std::round((u * 7.5 + 2.5) * avg_delay_ms_);
    }
    if (straggler_kind_[wid] == 1) {
      const double u =
          uniform01_host(cfg_.seed, (uint32_t)round_k, (uint32_t)wid);
      return std::round((u + 1.5) * avg_delay_ms_);
    }
    return 0.0;
  }

  DSCfg cfg_;
  at::Tensor w_, alpha_bar_;
  std::vector<c10::intrusive_ptr<c10d::ProcessGroup>> pgs_;
  std::vector<std::unique_ptr<Slot>> slots_;
  std::vector<std::thread> threads_;

  // LOCKING RULE: no pybind entry point may BLOCK on mu_ (or a Slot
  // mutex) while holding the GIL — C++ threads holding mu_ can need the
  // GIL transiently (destroying an at::Tensor whose Python wrapper exists
  // acquires the GIL for the pyobj decref), so a GIL-holding mu_ waiter
  // closes a deadlock cycle. Every binding that touches mu_/slots is
  // wrapped in py::gil_scoped_release; plain getters read fields lock-free.
  std::mutex mu_;
  std::condition_variable done_cv_;
  bool done_ = false;
  int64_t k_ = 0, clock_ = 0, applied_ = 0, rejected_ = 0;
  int64_t max_staleness_ = -1;
  int64_t elapsed_final_ms_ = 0;
  std::vector<uint8_t> avail_, last_accept_, dead_;
  int64_t threads_exited_ = 0;
  std::condition_variable threads_exit_cv_;
  std::string channel_error_;
  // bound the buffer like utils/trace.py's Tracer (long traced runs must
  // not grow without bound; ~2M events = a plottable trace)
  static constexpr size_t kTraceCap = 2'000'000;
  struct TraceEv {
    double ts;
    int32_t wid;
    int8_t kind;
    int32_t k, staleness;
  };
  std::vector<TraceEv> trace_ev_;
  std::deque<int> pending_;
  std::vector<double> submit_t_, finish_t_;
  std::vector<int64_t> waiting_ms_;
  std::set<int64_t> mark_at_;
  std::vector<int64_t> alpha_rows_;
  std::vector<at::Tensor> alpha_snap_;
  std::vector<uint8_t> alpha_ready_;
  std::condition_variable alpha_cv_;
  std::map<int64_t, double> marks_;
  std::vector<int64_t> opt_ms_;
  std::vector<at::Tensor> opt_w_;
  double t0_ = 0;
  // delay calibration
  double cul_time_ms_ = 0, avg_delay_ms_ = 0;
  int64_t cul_count_ = 0;
  bool delay_flag_ = false;
  std::vector<int> straggler_kind_;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  py::class_<DSCfg>(m, "DSCfg")
      .def(py::init<>())
      .def_readwrite("d", &DSCfg::d)
      .def_readwrite("P", &DSCfg::P)
      .def_readwrite("M", &DSCfg::M)
      .def_readwrite("N", &DSCfg::N)
      .def_readwrite("num_iter", &DSCfg::num_iter)
      .def_readwrite("printer_freq", &DSCfg::printer_freq)
      .def_readwrite("gamma", &DSCfg::gamma)
      .def_readwrite("batch_rate", &DSCfg::batch_rate)
      .def_readwrite("taw", &DSCfg::taw)
      .def_readwrite("gate", &DSCfg::gate)
      .def_readwrite("coeff", &DSCfg::coeff)
      .def_readwrite("seed", &DSCfg::seed)
      .def_readwrite("calib_window", &DSCfg::calib_window)
      .def_readwrite("asaga", &DSCfg::asaga)
      .def_readwrite("snapshot_weights", &DSCfg::snapshot_weights)
      .def_readwrite("k0", &DSCfg::k0)
      .def_readwrite("clock0", &DSCfg::clock0)
      .def_readwrite("bucket_ratio", &DSCfg::bucket_ratio)
      .def_readwrite("trace", &DSCfg::trace)
      .def_readwrite("worker_timeout_s", &DSCfg::worker_timeout_s);

  py::class_<DistServer>(m, "DistServer")
      .def(py::init<DSCfg, at::Tensor,
                    std::vector<c10::intrusive_ptr<c10d::ProcessGroup>>,
                    std::vector<int64_t>, std::vector<int64_t>>(),
           py::arg("cfg"), py::arg("w0"), py::arg("pair_pgs"),
           py::arg("mark_at"),
           py::arg("alpha_rows") = std::vector<int64_t>{})
      .def("start", &DistServer::start,
           py::call_guard<py::gil_scoped_release>())
      .def("wait_done", &DistServer::wait_done,
           py::call_guard<py::gil_scoped_release>())
      .def("join", &DistServer::join,
           py::call_guard<py::gil_scoped_release>())
      .def("local_next_dispatch", &DistServer::local_next_dispatch,
           py::call_guard<py::gil_scoped_release>())
      .def("local_deliver", &DistServer::local_deliver,
           py::call_guard<py::gil_scoped_release>())
      .def("k", &DistServer::k)
      .def("applied", &DistServer::applied)
      .def("rejected", &DistServer::rejected)
      .def("elapsed_ms", &DistServer::elapsed_ms)
      .def("waiting_ms", &DistServer::waiting_ms)
      .def("marks", &DistServer::marks)
      .def("opt_ms", &DistServer::opt_ms)
      .def("opt_w", &DistServer::opt_w)
      .def("weights", &DistServer::weights)
      .def("max_staleness_seen", &DistServer::max_staleness_seen)
      .def("dead_workers", &DistServer::dead_workers)
      .def("channel_error", &DistServer::channel_error)
      .def("trace_events", &DistServer::trace_events,
           py::call_guard<py::gil_scoped_release>())
      .def("delay_active", &DistServer::delay_active)
      .def("avg_delay_ms", &DistServer::avg_delay_ms)
      .def("clock", &DistServer::clock)
      .def("alpha_bar", &DistServer::alpha_bar)
      .def("request_alpha_snapshot", &DistServer::request_alpha_snapshot,
           py::call_guard<py::gil_scoped_release>())
      .def("wait_alpha", &DistServer::wait_alpha,
           py::call_guard<py::gil_scoped_release>())
      .def("get_alpha", &DistServer::get_alpha,
           py::call_guard<py::gil_scoped_release>())
      .def_static("delay_probe", &DistServer::delay_probe);
}
