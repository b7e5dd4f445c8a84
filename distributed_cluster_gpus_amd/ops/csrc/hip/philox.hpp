// Counter-based per-replica RNG for the batched engine (Philox4x32-10).
//
// Every Monte-Carlo replica owns an independent stream keyed on
// (seed, global_replica_id); the draw counter advances per call, so results
// are reproducible and independent of GPU count / kernel-launch chunking
// ("per-replica Philox counter-based RNG").
//
// Distribution recipes mirror the reference's sampling *distributions*
// (simcore/arrivals.py) — uniform(53-bit), exponential, Pareto, lognormal via
// Box-Muller.  GPU streams cannot be bitwise-equal to CPython's Mersenne
// (SURVEY §7 "RNG stream equivalence"); parity with the scalar engines is
// established distributionally (tests/test_gpu_engine.py).
#pragma once
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdint>

namespace dcg {

struct PhiloxState {
  uint64_t key;   // (seed ^ f(replica))
  uint64_t ctr;   // draw counter
};

__device__ __forceinline__ void philox_round(uint32_t& c0, uint32_t& c1,
                                             uint32_t& c2, uint32_t& c3,
                                             uint32_t k0, uint32_t k1) {
  constexpr uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t h0 = __umulhi(M0, c0), l0 = M0 * c0;
  uint32_t h1 = __umulhi(M1, c2), l1 = M1 * c2;
  uint32_t n0 = h1 ^ c1 ^ k0;
  uint32_t n1 = l1;
  uint32_t n2 = h0 ^ c3 ^ k1;
  uint32_t n3 = l0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

// One Philox4x32-10 block: 128 random bits from (key, ctr).
__device__ __forceinline__ void philox4x32(uint64_t key, uint64_t ctr,
                                           uint32_t out[4]) {
  uint32_t c0 = static_cast<uint32_t>(ctr);
  uint32_t c1 = static_cast<uint32_t>(ctr >> 32);
  uint32_t c2 = 0u, c3 = 0u;
  uint32_t k0 = static_cast<uint32_t>(key);
  uint32_t k1 = static_cast<uint32_t>(key >> 32);
  constexpr uint32_t B0 = 0x9E3779B9u, B1 = 0xBB67AE85u;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += B0; k1 += B1;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

// uniform double in [0,1) with 53-bit resolution (same combiner shape as
// CPython's genrand_res53, fed from Philox words).
__device__ __forceinline__ double u01(PhiloxState& st) {
  uint32_t w[4];
  philox4x32(st.key, st.ctr++, w);
  uint32_t a = w[0] >> 5, b = w[1] >> 6;
  return (a * 67108864.0 + b) * (1.0 / 9007199254740992.0);
}

// two independent uniforms from one block (for Box-Muller)
__device__ __forceinline__ void u01x2(PhiloxState& st, double& ua, double& ub) {
  uint32_t w[4];
  philox4x32(st.key, st.ctr++, w);
  ua = ((w[0] >> 5) * 67108864.0 + (w[1] >> 6)) * (1.0 / 9007199254740992.0);
  ub = ((w[2] >> 5) * 67108864.0 + (w[3] >> 6)) * (1.0 / 9007199254740992.0);
}

__device__ __forceinline__ double rexp(PhiloxState& st, double lambd) {
  return -log(1.0 - u01(st)) / lambd;
}

// integer in [0, n) — rejection-free (modulo bias negligible at n <= 8 but we
// use the widening-multiply trick for exactness up to 2^32)
__device__ __forceinline__ uint32_t rbelow(PhiloxState& st, uint32_t n) {
  uint32_t w[4];
  philox4x32(st.key, st.ctr++, w);
  return static_cast<uint32_t>((static_cast<uint64_t>(w[0]) * n) >> 32);
}

__device__ __forceinline__ double rnormal(PhiloxState& st, double mu, double sigma) {
  double ua, ub;
  u01x2(st, ua, ub);
  double r = sqrt(-2.0 * log(1.0 - ua));       // avoid log(0)
  double z = r * cos(2.0 * M_PI * ub);
  return mu + z * sigma;
}

__device__ __forceinline__ double rlognormal(PhiloxState& st, double mu, double sigma) {
  return exp(rnormal(st, mu, sigma));
}

// Pareto(xm=1, alpha) job-size draw (reference arrivals.py:5-9 shape)
__device__ __forceinline__ double rpareto_inf(PhiloxState& st) {
  double u = fmax(1e-9, 1.0 - u01(st));
  return 1.0 / pow(u, 1.0 / 1.8);
}

}  // namespace dcg
