// pybind11 bindings for the CDNA4 kernels (csrc/kernels.hip).
//
// The module is deliberately torch-ABI-free: functions take raw device
// pointers (as uintptr_t) + shapes + the HIP stream handle, which the thin
// Python adapter (asyncframework_amd/ops/hip.py) extracts from torch
// tensors. This keeps the extension a pure hipcc build (no hipify, no torch
// C++ headers) and lets the native runtime call the same launchers directly
// from C++. k_dev (nullable) is the device-resident round counter used by
// the hipGraph engine (engine/graph.py).

#include <pybind11/pybind11.h>

#include <hip/hip_runtime.h>
#include <cstdint>
#include <stdexcept>
#include <string>

namespace py = pybind11;

void register_libsvm(py::module_& m);  // csrc/libsvm_parser.cpp
void register_native_engine(py::module_& m);  // csrc/engine_native.cpp
void register_resident_engine(py::module_& m);  // csrc/engine_resident.hip

extern "C" {
void launch_grad_dense(const void*, const float*, const float*, float*,
                       float*, int*, const int*, long, int, uint64_t,
                       uint32_t, uint64_t, double, int, int, hipStream_t);
void launch_saga_grad_dense(const void*, const float*, const float*, float*,
                            float*, float*, int*, int*, float*, int*,
                            const int*, int, long, int, uint64_t, uint32_t,
                            uint64_t, double, int, int, hipStream_t);
int query_grad_grid(long);
void launch_reduce_partials(const float*, float*, int, int, int,
                            hipStream_t);
void launch_scan_rows(const float*, int*, float*, int*, const int*, long,
                      uint64_t, uint32_t, uint64_t, double, hipStream_t);
void launch_bump_counter(int*, hipStream_t);
void launch_grad_dense_list(const void*, const float*, float*, const int*,
                            const float*, const int*, long, int, int, int,
                            hipStream_t);
void launch_sgd_reduce_update(const float*, float*, int*, int*, float, float,
                              int, int, int, int, hipStream_t);
void launch_grad_csr(const int*, const int*, const void*, const float*,
                     const float*, float*, int*, const int*, long, uint64_t,
                     uint32_t, uint64_t, double, int, int, hipStream_t);
void launch_saga_grad_csr(const int*, const int*, const void*, const float*,
                          const float*, float*, float*, int*, int*, float*,
                          int*, const int*, int, long, uint64_t, uint32_t,
                          uint64_t, double, int, int, hipStream_t);
void launch_sgd_update(float*, const float*, float, float, int, hipStream_t);
void launch_saga_update(float*, const float*, float*, float, float, float,
                        int, hipStream_t);
void launch_saga_commit(float*, const int*, const float*, int, hipStream_t);
void launch_alpha_gather(float*, const float*, const int*, const int*, int,
                         hipStream_t);
void launch_sgd_update_fused(float*, float*, int*, float, float, int, int,
                             hipStream_t);
void launch_saga_update_fused(float*, float*, float*, int*, float, float,
                              float, int, hipStream_t);
}

static void check(hipError_t err, const char* what) {
  if (err != hipSuccess)
    throw std::runtime_error(std::string(what) + ": " +
                             hipGetErrorString(err));
}

PYBIND11_MODULE(_hip_core, m) {
  m.doc() = "MI355X CDNA4 kernels for asyncframework_amd";

  m.def("grad_dense",
        [](uintptr_t X, uintptr_t y, uintptr_t w, uintptr_t g,
           uintptr_t g_part, uintptr_t n, uintptr_t k_dev, long n_rows,
           int d, uint64_t seed, uint32_t round_k, uint64_t row_start,
           double rate, int objective, int x_is_bf16, uintptr_t stream) {
          launch_grad_dense((const void*)X, (const float*)y, (const float*)w,
                            (float*)g, (float*)g_part, (int*)n,
                            (const int*)k_dev, n_rows, d, seed, round_k,
                            row_start, rate, objective, x_is_bf16,
                            (hipStream_t)stream);
          check(hipGetLastError(), "grad_dense launch");
        });

  m.def("grad_grid", [](long n_rows) { return query_grad_grid(n_rows); });

  m.def("scan_rows",
        [](uintptr_t y, uintptr_t rowlist, uintptr_t ylist, uintptr_t count,
           uintptr_t scan_round, long n_rows, uint64_t seed, uint32_t round_k,
           uint64_t row_start, double rate, uintptr_t stream) {
          launch_scan_rows((const float*)y, (int*)rowlist, (float*)ylist,
                           (int*)count, (const int*)scan_round, n_rows, seed,
                           round_k, row_start, rate, (hipStream_t)stream);
          check(hipGetLastError(), "scan_rows launch");
        });

  m.def("sgd_reduce_update",
        [](uintptr_t g_part, uintptr_t w, uintptr_t k_dev, uintptr_t ticket,
           float gamma, float inv_batch, int num_part, int d, int G,
           int splits, uintptr_t stream) {
          launch_sgd_reduce_update((const float*)g_part, (float*)w,
                                   (int*)k_dev, (int*)ticket, gamma,
                                   inv_batch, num_part, d, G, splits,
                                   (hipStream_t)stream);
          check(hipGetLastError(), "sgd_reduce_update launch");
        });

  m.def("bump_counter", [](uintptr_t p, uintptr_t stream) {
    launch_bump_counter((int*)p, (hipStream_t)stream);
    check(hipGetLastError(), "bump_counter launch");
  });

  m.def("grad_dense_list",
        [](uintptr_t X, uintptr_t w, uintptr_t g_part, uintptr_t rowlist,
           uintptr_t ylist, uintptr_t count, long n_rows, int d,
           int objective, int x_is_bf16, uintptr_t stream) {
          launch_grad_dense_list((const void*)X, (const float*)w,
                                 (float*)g_part, (const int*)rowlist,
                                 (const float*)ylist, (const int*)count,
                                 n_rows, d, objective, x_is_bf16,
                                 (hipStream_t)stream);
          check(hipGetLastError(), "grad_dense_list launch");
        });

  m.def("reduce_partials",
        [](uintptr_t g_part, uintptr_t g, int d, int G, int splits,
           uintptr_t stream) {
          launch_reduce_partials((const float*)g_part, (float*)g, d, G,
                                 splits, (hipStream_t)stream);
          check(hipGetLastError(), "reduce_partials launch");
        });

  m.def("saga_grad_dense",
        [](uintptr_t X, uintptr_t y, uintptr_t w, uintptr_t alpha, uintptr_t g,
           uintptr_t g_part, uintptr_t n, uintptr_t idx, uintptr_t e,
           uintptr_t pos, uintptr_t k_dev, int commit_now, long n_rows,
           int d, uint64_t seed, uint32_t round_k, uint64_t row_start,
           double rate, int objective, int x_is_bf16, uintptr_t stream) {
          launch_saga_grad_dense((const void*)X, (const float*)y,
                                 (const float*)w, (float*)alpha, (float*)g,
                                 (float*)g_part, (int*)n, (int*)idx,
                                 (float*)e, (int*)pos, (const int*)k_dev,
                                 commit_now, n_rows, d, seed, round_k,
                                 row_start, rate, objective, x_is_bf16,
                                 (hipStream_t)stream);
          check(hipGetLastError(), "saga_grad_dense launch");
        });

  m.def("grad_csr",
        [](uintptr_t indptr, uintptr_t indices, uintptr_t values, uintptr_t y,
           uintptr_t w, uintptr_t g, uintptr_t n, uintptr_t k_dev,
           long n_rows, uint64_t seed, uint32_t round_k, uint64_t row_start,
           double rate, int objective, int v_is_bf16, uintptr_t stream) {
          launch_grad_csr((const int*)indptr, (const int*)indices,
                          (const void*)values, (const float*)y,
                          (const float*)w, (float*)g, (int*)n,
                          (const int*)k_dev, n_rows, seed, round_k, row_start,
                          rate, objective, v_is_bf16, (hipStream_t)stream);
          check(hipGetLastError(), "grad_csr launch");
        });

  m.def("saga_grad_csr",
        [](uintptr_t indptr, uintptr_t indices, uintptr_t values, uintptr_t y,
           uintptr_t w, uintptr_t alpha, uintptr_t g, uintptr_t n,
           uintptr_t idx, uintptr_t e, uintptr_t pos, uintptr_t k_dev,
           int commit_now, long n_rows, uint64_t seed, uint32_t round_k,
           uint64_t row_start, double rate, int objective, int v_is_bf16,
           uintptr_t stream) {
          launch_saga_grad_csr((const int*)indptr, (const int*)indices,
                               (const void*)values, (const float*)y,
                               (const float*)w, (float*)alpha, (float*)g,
                               (int*)n, (int*)idx, (float*)e, (int*)pos,
                               (const int*)k_dev, commit_now, n_rows, seed,
                               round_k, row_start, rate, objective, v_is_bf16,
                               (hipStream_t)stream);
          check(hipGetLastError(), "saga_grad_csr launch");
        });

  m.def("sgd_update", [](uintptr_t w, uintptr_t g, float gamma_k,
                         float inv_batch, int d, uintptr_t stream) {
    launch_sgd_update((float*)w, (const float*)g, gamma_k, inv_batch, d,
                      (hipStream_t)stream);
    check(hipGetLastError(), "sgd_update launch");
  });

  m.def("saga_update", [](uintptr_t w, uintptr_t g, uintptr_t ab, float gamma,
                          float inv_batch, float inv_N, int d,
                          uintptr_t stream) {
    launch_saga_update((float*)w, (const float*)g, (float*)ab, gamma,
                       inv_batch, inv_N, d, (hipStream_t)stream);
    check(hipGetLastError(), "saga_update launch");
  });

  m.def("saga_commit", [](uintptr_t alpha, uintptr_t idx, uintptr_t e, int n,
                          uintptr_t stream) {
    launch_saga_commit((float*)alpha, (const int*)idx, (const float*)e, n,
                       (hipStream_t)stream);
    check(hipGetLastError(), "saga_commit launch");
  });

  m.def("alpha_gather",
        [](uintptr_t a_dst, uintptr_t a_src, uintptr_t rows,
           uintptr_t count_dev, int cap, uintptr_t stream) {
          launch_alpha_gather((float*)a_dst, (const float*)a_src,
                              (const int*)rows, (const int*)count_dev, cap,
                              (hipStream_t)stream);
          check(hipGetLastError(), "alpha_gather launch");
        });

  m.def("sgd_update_fused",
        [](uintptr_t w, uintptr_t g, uintptr_t k_dev, float gamma,
           float inv_batch, int num_part, int d, uintptr_t stream) {
          launch_sgd_update_fused((float*)w, (float*)g, (int*)k_dev, gamma,
                                  inv_batch, num_part, d,
                                  (hipStream_t)stream);
          check(hipGetLastError(), "sgd_update_fused launch");
        });

  m.def("saga_update_fused",
        [](uintptr_t w, uintptr_t g, uintptr_t ab, uintptr_t k_dev,
           float gamma, float inv_batch, float inv_N, int d,
           uintptr_t stream) {
          launch_saga_update_fused((float*)w, (float*)g, (float*)ab,
                                   (int*)k_dev, gamma, inv_batch, inv_N, d,
                                   (hipStream_t)stream);
          check(hipGetLastError(), "saga_update_fused launch");
        });

  register_libsvm(m);
  register_native_engine(m);
  register_resident_engine(m);

  m.attr("__hip__") = true;
// source-provenance stamp: build_hip.py passes -DASYNCAMD_SRC_HASH=<sha256
// of csrc sources>; the Python loader refuses a binary whose stamp doesn't
// match the sources on disk (round-1 stale-binary postmortem).
#ifndef ASYNCAMD_SRC_HASH
#define ASYNCAMD_SRC_HASH "unstamped"
#endif
  m.attr("__src_hash__") = ASYNCAMD_SRC_HASH;
}
