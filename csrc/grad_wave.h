// Shared wave-dispatch descriptors between kernels.hip and the native
// engine (csrc/engine_native.cpp). One grad_dense_wave_kernel launch runs
// a whole quorum wave of dense-ASGD rounds: per-worker invariants live in
// a GradWaveSlot device table built once at engine init; per-round
// variables travel by value in GradWaveCmd (~1 KB, under the 4 KB kernel
// arg limit).
#pragma once

#define GRAD_WAVE_MAXP 64

struct GradWaveSlot {  // fixed per worker for the engine's lifetime
  const void* X;
  const float* y;
  const float* wbuf;
  float* g;
  int* n_out;
  long n_rows;
  long row_start;
  unsigned long long* done_flag;  // pinned host line (publish_done)
  unsigned long long* done_arr;   // device arrival counter
  // SAGA fields (null for ASGD)
  float* alpha;            // device table (resident) or staging (spill)
  int* idx_out;            // staged row list
  float* e_out;            // staged scalars
  int* pos_ctr;            // staging counter (wk.ctr + 4)
  const float* alpha_host; // pinned master table (spill mode) or null
  int* srows;              // spill refresh: sampled-row list
  float* sylist;           //   (y values — unused by the wave path)
  int* scnt;               //   sampled-row count
};

struct GradWaveCmd {  // per launch, by value
  int n;     // active slots in this wave
  int bper;  // blocks per worker (uniform across the wave)
  int interleave;  // 1: block b -> slot b%n (workers progress together);
                   // 0: block b -> slot b/bper (workers drain in order)
  int wid[GRAD_WAVE_MAXP];                  // slot index per wave member
  unsigned int round_k[GRAD_WAVE_MAXP];     // Philox round key (k_submit+1)
  unsigned long long done_val[GRAD_WAVE_MAXP];  // completion serial
};

// ---- CSR wave (SAGA + ASGD sparse) ----

struct CsrWaveSlot {  // fixed per worker for the engine's lifetime
  const int* indptr;
  const int* indices;
  const void* values;
  const float* y;
  const float* wbuf;
  float* alpha;      // worker-resident SAGA history slice (null for ASGD)
  float* g;
  int* n_out;        // wk.ctr
  int* idx_out;      // staging row list (SAGA)
  float* e_out;      // staging scalars (SAGA)
  int* pos_ctr;      // staging counter (wk.ctr + 4)
  long n_rows;
  long row_start;
  unsigned long long* done_flag;
  unsigned long long* done_arr;
};

struct CsrCommitCmd {  // per launch, by value: commit + staging reset pass
  int n;
  int bper;
  int wid[GRAD_WAVE_MAXP];
  int do_commit[GRAD_WAVE_MAXP];  // 1: scatter staged scalars, then reset;
                                  // 0: reset only (rejected previous round)
};

// unified commit view (dense SAGA — resident or host-spill — and CSR SAGA)
struct CommitSlot {
  float* dst;     // alpha_host (spill) or the worker's alpha table
  int* idx;       // staged row list
  float* e;       // staged scalars
  int* pos_ctr;   // staging counter to reset
  int* scnt;      // spill refresh counter to reset (null otherwise)
  int* n_out;     // sampled-count accumulator to reset
  unsigned long long* arr;  // per-slot arrival counter (reused done_arr)
};
