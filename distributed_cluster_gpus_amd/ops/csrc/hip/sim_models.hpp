// Device-side analytic models for the batched replica engine (gfx950).
//
// The DVFS power polynomial and T(n,f) latency model are fused into every
// step kernel as inlined device functions ("HIP device
// function, fused").  f64 forms are used on the simulation path (event times
// must be BITWISE-equal to the scalar engines; energy integrals need the
// precision).  The (n,f) grid search is candidate-strided over the replica's
// subgroup: at the default 64-lane width the 8x8 candidate grid maps exactly
// one-candidate-per-lane onto a CDNA4 wavefront; the
// 8-lane multi-replica build strides 8 candidates per lane.
#pragma once
#include <hip/hip_runtime.h>
#include <limits.h>

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

// ---- subwave reductions ----
// DCG_SUBWAVE lanes cooperate on one replica (64 = classic wave-per-replica;
// 8 = eight replicas per wavefront, amortizing the wave-uniform scalar work
// across subgroups).  xor-butterfly offsets < SUBW stay inside an aligned
// subgroup, so the same shfl_xor code serves both widths.
#ifndef DCG_SUBWAVE
#define DCG_SUBWAVE 64
#endif
constexpr int SUBW = DCG_SUBWAVE;
static_assert(SUBW == 64 || SUBW == 8, "supported subwave widths: 64, 8");

__device__ __forceinline__ int sub_lane() { return threadIdx.x & (SUBW - 1); }

__device__ __forceinline__ double wave_min_f64(double v) {
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1)
    v = fmin(v, __shfl_xor(v, off, 64));
  return v;
}

// argmin with lowest-absolute-lane tie-break within the subgroup: returns the
// min value; lane_out = the winning lane's ABSOLUTE in-wave index (valid as
// a __shfl source for the subgroup).
__device__ __forceinline__ double wave_argmin_f64(double v, int& lane_out) {
  int lane = threadIdx.x & 63;
  double bv = v;
  int bl = lane;
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1) {
    double ov = __shfl_xor(bv, off, 64);
    int ol = __shfl_xor(bl, off, 64);
    if (ov < bv || (ov == bv && ol < bl)) { bv = ov; bl = ol; }
  }
  lane_out = bl;
  return bv;
}

// generic (value, index) argmin: FIRST minimum in candidate-index order,
// matching the scalar engines' scan-order tie-break exactly.
__device__ __forceinline__ void wave_argmin_idx_f64(double v, int idx,
                                                    double& v_out, int& i_out) {
  double bv = v;
  int bi = idx;
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1) {
    double ov = __shfl_xor(bv, off, 64);
    int oi = __shfl_xor(bi, off, 64);
    if (ov < bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
  }
  v_out = bv;
  i_out = bi;
}

// (n, f) grid argmin over n_max*n_freq candidates, candidate-strided over the
// subgroup's lanes; candidate index ci = (n-1)*n_freq + f_idx reproduces the
// scalar scan order (n-major, frequency-minor, first-minimum tie-break) of
// policies/gridsearch.py::best_nf_grid.
// objective: 0 energy, 1 carbon (score=E*ci), 2 cost (score=E/3.6e6*price).
struct GridPick { int n; double f, T, P, E; bool found; };

__device__ __forceinline__ GridPick wave_grid_argmin(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n_max, int objective, double ci, double price,
    bool has_ddl, double ddl) {
  int lane = sub_lane();
  int n_cand = n_max * n_freq;
  double best_sc = D_INF;
  int best_ci = INT_MAX;
  for (int cand = lane; cand < n_cand; cand += SUBW) {
    int n = cand / n_freq + 1;
    double f = freq_levels[cand % n_freq];
    double T = d_unit_time(n, f, lc3);
    double E = d_job_power(n, f, pc3) * T;
    if (has_ddl && T > ddl) continue;
    double sc = E;
    if (objective == 1) sc = E * ci;
    else if (objective == 2) sc = (E / 3.6e6) * price;
    if (sc < best_sc) { best_sc = sc; best_ci = cand; }
  }
  double v;
  int wi;
  wave_argmin_idx_f64(best_sc, best_ci, v, wi);
  GridPick out;
  out.found = v < D_INF;
  if (!out.found) {
    out.n = 1; out.f = 0; out.T = 0; out.P = 0; out.E = 0;
    return out;
  }
  out.n = wi / n_freq + 1;
  out.f = freq_levels[wi % n_freq];
  out.T = d_unit_time(out.n, out.f, lc3);
  out.P = d_job_power(out.n, out.f, pc3);
  out.E = out.P * out.T;
  return out;
}

// energy-argmin over frequencies at fixed n (best_energy_freq),
// candidate-strided, first-minimum tie-break.
__device__ __forceinline__ double wave_energy_freq(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n) {
  int lane = sub_lane();
  double best_sc = D_INF;
  int best_k = INT_MAX;
  for (int k = lane; k < n_freq; k += SUBW) {
    double f = freq_levels[k];
    double sc = d_job_power(n, f, pc3) * d_unit_time(n, f, lc3);
    if (sc < best_sc) { best_sc = sc; best_k = k; }
  }
  double v;
  int wk;
  wave_argmin_idx_f64(best_sc, best_k, v, wk);
  return freq_levels[wk];
}

// ---- fp32 DECISION-SCORE variants (opt-in, BASELINE config 5's "fp16/fp32
// coeff eval") ----
// Only the candidate-ranking math drops to fp32; the chosen (n, f) feeds the
// usual f64 time/energy models, so event times and energy integrals keep
// full precision.  A decision can differ from the f64 path only when two
// candidates score within fp32 rounding of each other (documented,
// measurable divergence — off by default).
__device__ __forceinline__ float f_gpu_power(float f, const double* c3) {
  f = fmaxf(0.0f, f);
  return (float)c3[0] * f * f * f + (float)c3[1] * f + (float)c3[2];
}
__device__ __forceinline__ float f_unit_time(int n, float f, const double* c3) {
  n = max(1, n);
  f = fmaxf(1e-9f, f);
  if (n == 1) return (float)c3[0] + (float)c3[1] / f;
  return ((float)c3[0] + (float)c3[1] / f + (float)c3[2] * n) / n;
}

__device__ __forceinline__ GridPick wave_grid_argmin_f32(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n_max, int objective, double ci, double price,
    bool has_ddl, double ddl) {
  int lane = sub_lane();
  int n_cand = n_max * n_freq;
  float best_sc = 3.0e38f;
  int best_ci = INT_MAX;
  for (int cand = lane; cand < n_cand; cand += SUBW) {
    int n = cand / n_freq + 1;
    float f = (float)freq_levels[cand % n_freq];
    float T = f_unit_time(n, f, lc3);
    float E = (float)max(0, n) * f_gpu_power(f, pc3) * T;
    if (has_ddl && T > (float)ddl) continue;
    float sc = E;
    if (objective == 1) sc = E * (float)ci;
    else if (objective == 2) sc = (E / 3.6e6f) * (float)price;
    if (sc < best_sc) { best_sc = sc; best_ci = cand; }
  }
  double v;
  int wi;
  wave_argmin_idx_f64((double)best_sc, best_ci, v, wi);
  GridPick out;
  out.found = v < 3.0e38;
  if (!out.found) {
    out.n = 1; out.f = 0; out.T = 0; out.P = 0; out.E = 0;
    return out;
  }
  // full-precision models on the CHOSEN candidate
  out.n = wi / n_freq + 1;
  out.f = freq_levels[wi % n_freq];
  out.T = d_unit_time(out.n, out.f, lc3);
  out.P = d_job_power(out.n, out.f, pc3);
  out.E = out.P * out.T;
  return out;
}

__device__ __forceinline__ double wave_energy_freq_f32(
    const double* pc3, const double* lc3, const double* freq_levels,
    int n_freq, int n) {
  int lane = sub_lane();
  float best_sc = 3.0e38f;
  int best_k = INT_MAX;
  for (int k = lane; k < n_freq; k += SUBW) {
    float f = (float)freq_levels[k];
    float sc = (float)max(0, n) * f_gpu_power(f, pc3) * f_unit_time(n, f, lc3);
    if (sc < best_sc) { best_sc = sc; best_k = k; }
  }
  double v;
  int wk;
  wave_argmin_idx_f64((double)best_sc, best_k, v, wk);
  return freq_levels[wk];
}

}  // namespace dcg
