// Philox4x32-10 device implementation — MUST stay bit-identical to the
// numpy reference in asyncframework_amd/utils/philox.py (the shared-seed
// sampling contract: every worker/server derives the same per-row Bernoulli
// decision from (seed, round, absolute_row) with zero communication; the
// reference achieved this by re-running Spark's seeded BernoulliSampler,
// RandomSampler.scala:144, SparkASAGAThread.scala:372-376).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

// Full 4-word output for counter = row BLOCK (row/4): one eval decides four
// consecutive rows (word = row%4). Must match utils/philox.py bernoulli_mask.
__device__ __forceinline__ uint4 philox_block4(uint64_t seed, uint32_t round_k,
                                               uint64_t row_block) {
  uint32_t c0 = (uint32_t)(row_block & 0xFFFFFFFFull);
  uint32_t c1 = (uint32_t)(row_block >> 32);
  uint32_t c2 = round_k;
  uint32_t c3 = 0u;
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
#pragma unroll
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
  return make_uint4(c0, c1, c2, c3);
}

__device__ __forceinline__ uint32_t philox_x0(uint64_t seed, uint32_t round_k,
                                              uint64_t row) {
  uint32_t c0 = (uint32_t)(row & 0xFFFFFFFFull);
  uint32_t c1 = (uint32_t)(row >> 32);
  uint32_t c2 = round_k;
  uint32_t c3 = 0u;
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
#pragma unroll
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

// Host-side threshold: matches numpy's
// np.uint32(min(int(rate * 2**32), 2**32-1)).
static inline uint32_t philox_threshold(double rate) {
  if (rate >= 1.0) return 0xFFFFFFFFu;  // mask code treats rate>=1 as all-in
  double t = rate * 4294967296.0;
  if (t < 0) t = 0;
  uint64_t ti = (uint64_t)t;
  return (uint32_t)(ti > 0xFFFFFFFFull ? 0xFFFFFFFFull : ti);
}
