// Device-side analytic models for the batched replica engine (gfx950).
//
// The DVFS power polynomial and T(n,f) latency model are fused into every
// step kernel as inlined device functions ("HIP device
// function, fused").  f64 forms are used on the simulation path (energy
// integrals and event times need the precision); the wave-parallel grid
// search evaluates all 64 (n,f) candidates with one lane each — the 8x8
// candidate grid maps exactly onto CDNA4's 64-wide wavefront
// (SURVEY §2 row 8).
#pragma once
#include <hip/hip_runtime.h>

namespace dcg {

constexpr double D_INF = 1e300;

__device__ __forceinline__ double d_gpu_power(double f, const double* c3) {
  f = fmax(0.0, f);
  return c3[0] * f * f * f + c3[1] * f + c3[2];
}
__device__ __forceinline__ double d_job_power(int n, double f, const double* c3) {
  return max(0, n) * d_gpu_power(f, c3);
}
__device__ __forceinline__ double d_unit_time(int n, double f, const double* c3) {
  n = max(1, n);
  f = fmax(1e-9, f);
  if (n == 1) return c3[0] + c3[1] / f;
  return (c3[0] + c3[1] / f + c3[2] * n) / n;
}

// ---- wavefront reductions (64-wide; __shfl_xor over all 64 lanes) ----
__device__ __forceinline__ double wave_min_f64(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmin(v, __shfl_xor(v, off, 64));
  return v;
}

// argmin with lowest-lane tie-break: returns min value; *lane_out = winner.
__device__ __forceinline__ double wave_argmin_f64(double v, int& lane_out) {
  int lane = threadIdx.x & 63;
  double bv = v;
  int bl = lane;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    double ov = __shfl_xor(bv, off, 64);
    int ol = __shfl_xor(bl, off, 64);
    if (ov < bv || (ov == bv && ol < bl)) { bv = ov; bl = ol; }
  }
  lane_out = bl;
  return bv;
}

// 64-candidate (n, f) grid argmin.  lane = (n-1)*n_freq + f_idx reproduces the
// scalar scan order (n-major, frequency-minor, first-minimum tie-break) of
// policies/gridsearch.py::best_nf_grid.  Lanes beyond the grid contribute +inf.
// objective: 0 energy, 1 carbon (score=E*ci), 2 cost (score=E/3.6e6*price).
struct GridPick { int n; double f, T, P, E; bool found; };

__device__ __forceinline__ GridPick wave_grid_argmin(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n_max, int objective, double ci, double price,
    bool has_ddl, double ddl) {
  int lane = threadIdx.x & 63;
  int n = lane / n_freq + 1;
  int fi = lane % n_freq;
  double score = D_INF, T = 0, P = 0, E = 0, f = 0;
  if (lane < n_max * n_freq) {
    f = freq_levels[fi];
    T = d_unit_time(n, f, lc3);
    P = d_job_power(n, f, pc3);
    E = P * T;
    if (!(has_ddl && T > ddl)) {
      score = E;
      if (objective == 1) score = E * ci;
      else if (objective == 2) score = (E / 3.6e6) * price;
    }
  }
  int wl;
  double best = wave_argmin_f64(score, wl);
  GridPick out;
  out.found = best < D_INF;
  out.n = __shfl(n, wl, 64);
  out.f = __shfl(f, wl, 64);
  out.T = __shfl(T, wl, 64);
  out.P = __shfl(P, wl, 64);
  out.E = __shfl(E, wl, 64);
  return out;
}

// energy-argmin over frequencies at fixed n (best_energy_freq): lane-parallel
// over n_freq lanes, first-minimum tie-break.
__device__ __forceinline__ double wave_energy_freq(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n) {
  int lane = threadIdx.x & 63;
  double score = D_INF, f = 0;
  if (lane < n_freq) {
    f = freq_levels[lane];
    score = d_job_power(n, f, pc3) * d_unit_time(n, f, lc3);
  }
  int wl;
  wave_argmin_f64(score, wl);
  return __shfl(f, wl, 64);
}

}  // namespace dcg
