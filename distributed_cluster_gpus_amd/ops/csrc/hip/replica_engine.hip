// Batched Monte-Carlo replica engine for MI355X (gfx950, CDNA4).
//
// Architecture (SURVEY §7 "fixed-Δt lockstep engine with exact event times"):
// each CDNA4 wavefront (64 lanes) owns ONE replica of the multi-DC simulation
// and advances it event-by-event with exact event times; thousands of
// replicas run concurrently (256 CUs x 4 SIMDs x N waves).  All per-replica
// state lives in HBM3E as replica-major structure-of-arrays, so lane-strided
// scans (job-slot min-finish, empty-slot search, queue walks) are coalesced.
// Next-event selection is a wave argmin over {16 arrival streams, in-flight
// WAN transfers, per-DC cached min finish times, next log tick}.  The
// (n, f) grid search maps the full 8x8 candidate grid onto the 64 lanes of
// one wavefront (sim_models.hpp::wave_grid_argmin).
//
// Execution discipline: control flow is wave-uniform — every lane computes
// the same scalar values redundantly; global-memory writes are guarded to
// lane 0; cooperative phases (scans/argmins) use lane-strided reads + shfl
// reductions.  Replicas are independent, so kernels need no inter-workgroup
// communication; multi-GPU scaling shards replicas per rank
// (parallel/sharding.py) with RCCL used only for metric reductions.
//
// Event semantics mirror the scalar engines (engine/oracle.py ==
// reference simulator simcore/simulator_paper_multi.py:412-480): per-event
// energy/util accrual using per-job f_used power, queue-drain on completion
// with inference priority, per-algorithm admission decisions, the
// finish%log_interval job-unit remainder quirk, the end-of-run baseline-model
// energy flush.  RNG is Philox (philox.hpp) — parity with the scalar engines
// is distributional, not bitwise (SURVEY §7 "RNG stream equivalence").
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <vector>

#include "philox.hpp"
#include "sim_models.hpp"

namespace dcg {

constexpr int MAX_DC = 8;
constexpr int MAX_FREQ = 8;

// Packed per-object payload records: everything a job start/finish (or a
// WAN-transfer push/admission) touches lands in ONE cache line instead of
// 6-12 scattered SoA arrays — the engine is memory-LATENCY-chain bound
// (profiles/README.md PMC analysis), so dependent line count per event is
// the cost that matters.  Scan-heavy fields (finish times, gpus, jtype)
// stay SoA/LDS.
struct XRec {            // 32 B: in-flight WAN transfer payload
  double size;
  float netlat;
  int jid;
  short nsel;            // RL-chosen n (chsac; unclamped g+1)
  unsigned char dc, jtype, ing, adc, ag, mdc;
  unsigned char mg, has_rl, pad0, pad1;
};
static_assert(sizeof(XRec) == 32, "XRec must be one 32B record");

struct SRec {            // 64 B: cold job-slot payload
  double start, lastupd, size, fused, done;
  float netlat;
  int jid;
  int seq;
  unsigned char ing, pcount, nrew, adc, ag, mdc, mg, has_rl;
  unsigned char pad[4];
};
static_assert(sizeof(SRec) == 64, "SRec must be one 64B record");

enum Algo { A_DEFAULT = 0, A_CAP_UNIFORM, A_CAP_GREEDY, A_JOINT_NF, A_BANDIT,
            A_CARBON_COST, A_ECO_ROUTE, A_DEBUG, A_CHSAC };

// CHSAC-AF action-request state machine kinds
enum PendKind : int { PEND_NONE = 0, PEND_ARRIVAL = 1, PEND_DRAIN = 2,
                      PEND_REALLOC = 3 };
// request flags
enum ReqFlag : int { REQ_IDLE = 0, REQ_PENDING = 1, REQ_READY = 2 };

constexpr int LAT_BINS = 64;  // log10 bins over sojourn [1e-4 s, 1e4 s)

enum Err : int { ERR_NONE = 0, ERR_QUEUE_OVF = 1, ERR_XFER_OVF = 2,
                 ERR_SLOT_OVF = 4, ERR_LOG_OVF = 8 };

struct EngineDesc {
  // sizes
  int n_rep, n_dc, n_ing, n_freq, n_streams, total_slots, tcap, qcap;
  double end_time, log_interval;
  // scenario constants (device)
  const double* freq_levels;      // [n_freq]
  const double* pc;               // [dc][2][3]
  const double* lc;               // [dc][2][3]
  const double* wan_lat;          // [ing][dc]
  const double* wan_bw;           // [ing][dc]
  const double* carbon;           // [dc]
  const double* price24;          // [24]
  const int* total_gpus;          // [dc]
  const double* p_idle;           // [dc]
  const double* p_sleep;          // [dc]
  const double* p_peak;           // [dc] (baseline end-flush model)
  const double* pow_alpha;        // [dc]
  const int* power_gating;        // [dc]
  const double* default_freq;     // [dc]
  const int* slot_off;            // [dc+1]
  const int* slot_dc;             // [total_slots] slot -> dc
  // policy / run params
  int algo, max_gpj, inf_priority, scale_out_low, energy_aware;
  double dvfs_low, dvfs_high, power_cap;
  int eco_obj, num_fixed;
  int fp32_score;                 // opt-in fp32 decision-score eval (grid /
                                  //   energy-freq argmins; times stay f64)
  double fixed_freq;
  double payload_gb[2];
  // arrival processes by jtype: mode 0=poisson 1=sinusoid 2=off
  int arr_mode[2];
  double arr_rate[2], arr_amp[2], arr_period[2];
  uint64_t seed;
  int64_t rep_id_offset;          // global replica id of local replica 0
  // per-replica scalars
  double* now;                    // [r] (last event time; <0 = no event yet)
  double* next_log;               // [r]
  uint64_t* rng_ctr;              // [r]
  int* jid_ctr;                   // [r]
  int* done;                      // [r]
  int* err;                       // [r]
  double* arr_next;               // [r][n_streams]  (stream = ing*2 + jtype)
  // per (r, dc)
  int* busy;
  double* cur_freq;
  double* energy_j;
  double* util_time;
  double* util_begin;             // [r][dc] first-event stamp (<0 unset)
  double* acc_unit;
  double* p_active;               // cached sum of running-job powers
  double* sum_tpt;                // cached sum of running-job throughputs
  int* n_running;
  double* dc_min_finish;          // cached min finish time (INF if none)
  int* dc_min_slot;
  // job slots [r][total_slots]: finish times + scan fields SoA, the cold
  // payload as one SRec per slot (f64 values bitwise-exact as before)
  double* s_finish;               // INF = empty
  SRec* s_rec;                    // packed payload
  short* s_gpus;                  // 0 = empty (dense busy/log scans)
  char* s_jtype;                  // dense scans (log rows, elastic, cap)
  // in-flight transfers [r][tcap]: time SoA (LDS-mirrored), payload packed
  double* x_time;                 // INF = empty
  XRec* x_rec;
  // queues [r][dc][2] ring of capacity qcap; per-entry payload = (size,
  // enqueue-time) f64 pair; aux fields (netlat/jid/ing) exist only for the
  // logging replica ([dc][2][qcap], no replica dim)
  int* q_head;
  int* q_len;
  double* q_pay;                  // [r][dc][2][qcap][2] (f64 exact parity)
  float* q_netlat;                // [dc][2][qcap] (log replica only)
  int* q_jid;                     // [dc][2][qcap] (log replica only)
  char* q_ing;                    // [dc][2][qcap] (log replica only)
  // bandit state [r][dc][2][n_freq]
  int* b_n;
  double* b_s;
  long long* b_t;                 // [r]
  // metrics [r]
  long long* ev_count;
  long long* jobs_done;
  long long* jobs_done_inf;
  double* sum_lat;
  double* sum_lat_inf;
  double* sum_wait;               // summed queueing delay (job start minus
                                  //   its enqueue at the DC; 0 for jobs that
                                  //   start straight off the WAN transfer)
  int* seq_ctr;                   // [r] job-start sequence counter
  double* snap_f;                 // [r][total_slots] cap_greedy pass snapshot
                                  //   (allocated only for algo==cap_greedy)
  // logging (one designated replica); the host drains both buffers
  // chunk-wise between launches, so caps bound one launch's production only
  int log_replica;                // -1 = off
  int cl_cap, jl_cap;
  int* cl_count;                  // [1]
  double* cl_rows;                // [cl_cap][15]
  int* jl_count;                  // [1]
  double* jl_rows;                // [jl_cap][11]
  // ===== arrival-trace replay mode (single-replica exact-parity testing:
  // arrivals come from a host-recorded (time, size) FIFO per stream instead
  // of the Philox draws) =====
  int trace_mode;                 // 0 off, 1 on
  int trace_cap;                  // entries per stream
  const double* trace_time;       // [r][n_streams][cap] absolute times
  const double* trace_size;       // [r][n_streams][cap]
  const char* trace_dc;           // [r][n_streams][cap] routed DC (-1 = use
                                  //   the algorithm's own routing)
  int* trace_pos;                 // [r][n_streams] cursor
  // ===== CHSAC-AF (RL-in-the-loop) state; null unless algo == A_CHSAC =====
  int obs_dim;                    // 1 + 6*n_dc
  double sla_p99_ms;
  // --- device-side policy serving (serve_device=1): the actor MLP runs
  // INSIDE the advance kernel from a flat weights buffer the host refreshes
  // after each SAC train round, so replicas never pause for a host policy
  // round-trip (round-1 bottleneck: 206k ev/s host-loop-bound).  Layout of
  // pw (all fp32, [in][out]-major so the out index is lane-coalesced):
  //   enc1 Wt[obs_dim][hid] b[hid] | enc2 Wt[hid][hid] b | enc3 Wt[hid][hid] b
  //   | hdc1 Wt[hid][hid] b | hdc2 Wt[hid][n_dc] b[n_dc]
  //   | hg1 Wt[hid][hid] b | hg2 Wt[hid][n_g] b[n_g]
  int serve_device;               // 0 = host pause/resume, 1 = in-kernel
  int hid;                        // encoder/head hidden width (<= RL_MAX_HID)
  int n_g;                        // g-head size (max_gpj)
  int rl_det;                     // 1 = greedy argmax (no Gumbel draw)
  int tr_limit;                   // yield the launch when the transition ring
                                  //   reaches this (bounds policy staleness)
  const float* pw;                // flat policy weights
  // action request/response, one slot per replica
  int* req_flag;                  // [r] ReqFlag
  float* req_obs;                 // [r][obs_dim]
  int* req_mdc;                   // [r] bitmask of valid DCs
  int* req_mg;                    // [r] bitmask of valid g choices
  int* resp_dc;                   // [r]
  int* resp_g;                    // [r]
  int* pend_kind;                 // [r] PendKind
  // stashed context for the paused event
  double* pend_size;              // [r]
  float* pend_netlat;             // [r]
  int* pend_jid;                  // [r]
  int* pend_ing;                  // [r]
  int* pend_jt;                   // [r]
  int* pend_dc;                   // [r] (drain: source DC)
  int* pend_from_inf;             // [r] (drain: which queue the job came from)
  double* pend_enq;               // [r] (drain: the popped job's enqueue time)
  // per-job RL obs traces (the action/mask bytes live in SRec/XRec;
  // arrival path keeps nrew = g_idx+1 unclamped, drain path clamped —
  // reference :571 vs :889)
  float* slot_s0;                 // [r][total_slots][obs_dim]
  float* x_s0;                    // [r][tcap][obs_dim]
  // elastic scaling (chsac only): preempted-training-job pool per replica
  int elastic;                    // 0 off, 1 on
  int pp_cap;                     // pool capacity
  int* pp_count;                  // [r] entries in pool
  int* pp_cursor;                 // [r] next entry to reallocate
  double* pp_size;                // [r][cap] original size
  double* pp_done;                // [r][cap] units completed at preemption
  double* pp_start;               // [r][cap] ORIGINAL start time (resume
                                  //   keeps it: latency spans preemptions)
  float* pp_netlat;               // [r][cap]
  int* pp_jid;                    // [r][cap]
  unsigned char* pp_ing;          // [r][cap]
  unsigned char* pp_dc;           // [r][cap]
  unsigned char* pp_pcount;       // [r][cap] job's preempt count
  float* pp_s0;                   // [r][cap][obs_dim] carried RL trace
  unsigned char* pp_adc;          // [r][cap]
  unsigned char* pp_ag;           // [r][cap]
  unsigned char* pp_nrew;         // [r][cap]
  unsigned char* pp_has_rl;       // [r][cap]
  // latency histograms per (r, jtype): counts in log10 bins
  int* lat_hist;                  // [r][2][LAT_BINS]
  long long* lat_count;           // [r][2]
  double* lat_sum;                // [r][2]
  // exact sliding-window p99 (parity mode): the reference computes an exact
  // percentile over the last 2048 sojourns (simulator_paper_multi.py:728-737)
  // where the fast path approximates from the log histogram.  exact_p99=1
  // maintains a SORTED window + an insertion-order ring per (r, jtype) and
  // reproduces np.percentile(buf, 99) bit-for-bit.
  int exact_p99;
  int p99_win;                    // window size (2048)
  double* p99_sorted;             // [r][2][win]
  double* p99_ring;               // [r][2][win]
  // transition ring (global across replicas; host drains between launches)
  int tr_cap;
  int* tr_count;                  // [1] atomicAdd cursor
  float* tr_s0;                   // [cap][obs_dim]
  float* tr_s1;                   // [cap][obs_dim]
  unsigned char* tr_adc;          // [cap]
  unsigned char* tr_ag;           // [cap]
  float* tr_r;                    // [cap]
  float* tr_costs;                // [cap][3]: latency_p99_ms, power_W, gpu_over
  unsigned char* tr_mdc;          // [cap]
  unsigned char* tr_mg;           // [cap]
};

// ---------------- wave helpers ----------------
__device__ __forceinline__ int wave_sum_i32(int v) {
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ void store_fence() {
  // make lane-0 stores (global AND LDS) visible to this wave's subsequent
  // loads: full hardware wait + compiler reordering barrier
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
}

__device__ __forceinline__ void lds_fence() {
  // lighter fence for sites that only wrote LDS: waiting lgkmcnt alone
  // avoids draining unrelated outstanding global loads (vmcnt)
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

// ---------------- hot per-replica state (LDS-resident) ----------------
// The fields every event touches live in LDS for the kernel's lifetime
// (~850 B per replica, ~3.4 KB per 4-replica workgroup of the CU's 160 KiB)
// and are written back to HBM at kernel exit.  This removes most of the
// dependent global-memory round trips per event (PMC before: waves waiting
// 63% of cycles on memory; see profiles/README.md).
constexpr int MAX_STREAMS = 16;
constexpr int THREADS_PER_BLOCK = 128;
constexpr int REPLICAS_PER_BLOCK = THREADS_PER_BLOCK / DCG_SUBWAVE;
struct Hot {
  double arr_next[MAX_STREAMS];
  // per-replica scalars updated every arrival/finish — LDS-resident so the
  // hot path never does global read-modify-writes for them
  double sum_lat, sum_lat_inf, sum_wait;
  long long jobs_done, jobs_done_inf;
  int jid_ctr, seq_ctr;
  double dc_minf[MAX_DC];
  double p_active[MAX_DC];
  double sum_tpt[MAX_DC];
  double energy[MAX_DC];
  double util_time[MAX_DC];
  double acc_unit[MAX_DC];
  double util_begin[MAX_DC];
  double next_log;
  double cur_freq[MAX_DC];
  int dc_mins[MAX_DC];
  int busy[MAX_DC];
  int n_running[MAX_DC];
  int q_len[MAX_DC * 2];
  int q_head[MAX_DC * 2];
};

// ---------------- replica context ----------------
struct Ctx {
  const EngineDesc* S;
  Hot* hs;      // LDS-resident hot state of this wave's replica
  double* l_fin;  // LDS mirror of this replica's s_finish[total_slots]
  double* l_xt;   // LDS mirror of this replica's x_time[tcap]
  // chsac serve-device scratch (LDS): obs vector, two activation ping-pong
  // buffers, and the head logits (dc at +0, g at +8)
  float* l_obs;
  float* l_act_a;
  float* l_act_b;
  float* l_logits;
  int r;        // local replica index
  int lane;
  PhiloxState rng;
  double now;       // last processed event time (or -1)
  double t_first;   // first event time (-1 until known)

  __device__ const double* pc3(int d, int jt) const { return &S->pc[(d * 2 + jt) * 3]; }
  __device__ const double* lc3(int d, int jt) const { return &S->lc[(d * 2 + jt) * 3]; }
  __device__ int free_gpus(int d) const {
    return S->total_gpus[d] - hs->busy[d];
  }
  __device__ double dc_power(int d) const {
    int idle = free_gpus(d);
    double pi = S->power_gating[d] ? S->p_sleep[d] : S->p_idle[d];
    return hs->p_active[d] + idle * pi;
  }
  __device__ double price_kwh(double t) const {
    int h = static_cast<int>(fmod(t, 86400.0) / 3600.0);
    return S->price24[h];
  }
};

// recompute a DC's cached min finish (wave-cooperative)
__device__ void rescan_dc_min(Ctx& c, int d) {
  const EngineDesc& S = *c.S;
  int lo = S.slot_off[d], hi = S.slot_off[d + 1];
  double v = D_INF;
  int slot = -1;
  for (int k = lo + c.lane; k < hi; k += SUBW) {
    double f = c.l_fin[k];
    if (f < v) { v = f; slot = k; }
  }
  int wl;
  double best = wave_argmin_f64(v, wl);
  slot = __shfl(slot, wl, 64);
  if (c.lane == 0) {
    c.hs->dc_minf[d] = best;
    c.hs->dc_mins[d] = best < D_INF ? slot : -1;
  }
  lds_fence();
}

// start a job on DC d with (n, f); assumes free >= 1; wave-cooperative.
__device__ void start_job(Ctx& c, int d, int jt, double size, float netlat,
                          int jid, int ing, int n, double f, double now) {
  const EngineDesc& S = *c.S;
  int64_t base = (int64_t)c.r * S.total_slots;
  int lo = S.slot_off[d], hi = S.slot_off[d + 1];
  // find first empty slot (finish == INF <=> s_gpus == 0), from the LDS mirror
  int cand = INT_MAX;
  for (int k = lo + c.lane; k < hi; k += SUBW) {
    if (c.l_fin[k] >= D_INF) { cand = k; break; }
  }
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1)
    cand = min(cand, __shfl_xor(cand, off, 64));
  if (cand == INT_MAX) {  // cannot happen if capacities == total_gpus
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_SLOT_OVF);
    return;
  }
  double T = d_unit_time(n, f, c.lc3(d, jt));
  double finish = now + (double)size * T;
  if (c.lane == 0) {
    c.l_fin[cand] = finish;
    SRec* sr = &S.s_rec[base + cand];
    sr->start = now;
    sr->lastupd = now;
    sr->size = size;
    sr->fused = f;
    sr->netlat = netlat;
    sr->jid = jid;
    sr->done = 0.0;
    sr->pcount = 0;
    sr->ing = (unsigned char)ing;
    sr->seq = ++c.hs->seq_ctr;
    sr->has_rl = 0;
    S.s_gpus[base + cand] = (short)n;
    S.s_jtype[base + cand] = (char)jt;
    c.hs->busy[d] += n;
    c.hs->n_running[d] += 1;
    c.hs->p_active[d] += d_job_power(n, f, c.pc3(d, jt));
    c.hs->sum_tpt[d] += 1.0 / T;
    if (finish < c.hs->dc_minf[d]) {
      c.hs->dc_minf[d] = finish;
      c.hs->dc_mins[d] = cand;
    }
  }
  store_fence();
}

// heuristic allocator (policy.py:16-41 semantics); mutates cur_freq; returns g
__device__ int heuristic_alloc(Ctx& c, int d, int jt) {
  const EngineDesc& S = *c.S;
  int free = c.free_gpus(d);
  int g = free > 0 ? min(free, S.max_gpj) : 0;
  double cf = c.hs->cur_freq[d];
  double nf = cf;
  if (!S.energy_aware) {  // perf_first
    if (jt == 0) nf = S.dvfs_high;
    else {
      int qi = c.hs->q_len[d * 2 + 0];
      nf = fmax(cf, qi > 0 ? S.dvfs_high : S.default_freq[d]);
    }
  } else {
    if (jt == 0) nf = S.dvfs_high;
    else if (S.scale_out_low && free >= 2) {
      nf = S.dvfs_low;
      g = min(free, S.max_gpj);
    } else nf = fmax(cf, S.dvfs_low);
  }
  if (c.lane == 0) c.hs->cur_freq[d] = nf;
  lds_fence();
  return max(1, g);
}

// per-algorithm (n, f) decision at admission; returns chosen n, f
template <int ALGO>
__device__ void decide_nf(Ctx& c, int d, int jt, float size, double now,
                          int& n_out, double& f_out) {
  const EngineDesc& S = *c.S;
  int free = c.free_gpus(d);
  if (ALGO == A_JOINT_NF) {
    GridPick g = S.fp32_score
        ? wave_grid_argmin_f32(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                               S.n_freq, S.max_gpj, 0, 0.0, 0.0, false, 0.0)
        : wave_grid_argmin(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                           S.n_freq, S.max_gpj, 0, 0.0, 0.0, false, 0.0);
    n_out = g.n; f_out = g.f;
  } else if (ALGO == A_BANDIT) {
    // UCB1 (learners.py:20-36): uniform-redundant serial loop over arms
    long long t = S.b_t[c.r] + 1;
    if (c.lane == 0) S.b_t[c.r] = t;
    int64_t ab = ((int64_t)c.r * S.n_dc + d) * 2 + jt;
    double bf = -1.0;
    for (int k = 0; k < S.n_freq; ++k)
      if (S.b_n[ab * S.n_freq + k] < 1) { bf = S.freq_levels[k]; break; }
    if (bf < 0) {
      double best_ucb = -1e9;
      for (int k = 0; k < S.n_freq; ++k) {
        int nn = S.b_n[ab * S.n_freq + k];
        double mean = S.b_s[ab * S.n_freq + k] / nn;
        double ucb = mean + sqrt(2.0 * log((double)t) / nn);
        if (ucb > best_ucb) { best_ucb = ucb; bf = S.freq_levels[k]; }
      }
    }
    n_out = min(free, S.max_gpj);
    f_out = bf;
  } else if (ALGO == A_CARBON_COST) {
    double price = c.price_kwh(now);
    int obj = price > 0.0 ? 2 : 1;
    double ci = price > 0.0 ? 0.0 : S.carbon[d];
    GridPick g = S.fp32_score
        ? wave_grid_argmin_f32(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                               S.n_freq, S.max_gpj, obj, ci, price, false, 0.0)
        : wave_grid_argmin(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                           S.n_freq, S.max_gpj, obj, ci, price, false, 0.0);
    n_out = g.n; f_out = g.f;
  } else if (ALGO == A_DEBUG) {
    n_out = S.num_fixed;
    f_out = S.fixed_freq > 0 ? S.fixed_freq
            : (S.fp32_score
               ? wave_energy_freq_f32(c.pc3(d, jt), c.lc3(d, jt),
                                      S.freq_levels, S.n_freq, S.num_fixed)
               : wave_energy_freq(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                                  S.n_freq, S.num_fixed));
  } else {  // heuristic family
    n_out = heuristic_alloc(c, d, jt);
    f_out = c.hs->cur_freq[d];
  }
}

// queue ops (wave-uniform; lane-0 writes).  `enq` is the job's enqueue time
// at this DC — carried per entry so pops can credit the queueing delay.
__device__ bool queue_push(Ctx& c, int d, int jt, double size, float netlat,
                           int jid, int ing, double enq) {
  const EngineDesc& S = *c.S;
  int ql = d * 2 + jt;
  int len = c.hs->q_len[ql];
  if (len >= S.qcap) {
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_QUEUE_OVF);
    return false;
  }
  int pos = (c.hs->q_head[ql] + len) % S.qcap;
  if (c.lane == 0) {
    int64_t at = ((int64_t)(c.r * S.n_dc + d) * 2 + jt) * S.qcap + pos;
    S.q_pay[at * 2] = size;
    S.q_pay[at * 2 + 1] = enq;
    if (c.r == S.log_replica) {
      int64_t aux = ((int64_t)ql) * S.qcap + pos;
      S.q_netlat[aux] = netlat;
      S.q_jid[aux] = jid;
      S.q_ing[aux] = (char)ing;
    }
    c.hs->q_len[ql] = len + 1;
  }
  store_fence();
  return true;
}

__device__ bool queue_push_front(Ctx& c, int d, int jt, double size,
                                 float netlat, int jid, int ing, double enq) {
  const EngineDesc& S = *c.S;
  int ql = d * 2 + jt;
  int len = c.hs->q_len[ql];
  if (len >= S.qcap) {
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_QUEUE_OVF);
    return false;
  }
  int pos = (c.hs->q_head[ql] - 1 + S.qcap) % S.qcap;
  if (c.lane == 0) {
    int64_t at = ((int64_t)(c.r * S.n_dc + d) * 2 + jt) * S.qcap + pos;
    S.q_pay[at * 2] = size;
    S.q_pay[at * 2 + 1] = enq;
    if (c.r == S.log_replica) {
      int64_t aux = ((int64_t)ql) * S.qcap + pos;
      S.q_netlat[aux] = netlat;
      S.q_jid[aux] = jid;
      S.q_ing[aux] = (char)ing;
    }
    c.hs->q_head[ql] = pos;
    c.hs->q_len[ql] = len + 1;
  }
  store_fence();
  return true;
}

__device__ bool queue_pop(Ctx& c, int d, int jt, double& size, float& netlat,
                          int& jid, int& ing, double& enq) {
  const EngineDesc& S = *c.S;
  int ql = d * 2 + jt;
  int len = c.hs->q_len[ql];
  if (len <= 0) return false;
  int pos = c.hs->q_head[ql];
  int64_t at = ((int64_t)(c.r * S.n_dc + d) * 2 + jt) * S.qcap + pos;
  size = S.q_pay[at * 2];
  enq = S.q_pay[at * 2 + 1];
  if (c.r == S.log_replica) {
    int64_t aux = ((int64_t)ql) * S.qcap + pos;
    netlat = S.q_netlat[aux];
    jid = S.q_jid[aux];
    ing = S.q_ing[aux];
  } else {
    netlat = 0.0f; jid = 0; ing = 0;
  }
  if (c.lane == 0) {
    c.hs->q_head[ql] = (pos + 1) % S.qcap;
    c.hs->q_len[ql] = len - 1;
  }
  store_fence();
  return true;
}

// drain queues after a completion (inference first; reference :839-927)
template <int ALGO>
__device__ void drain_queues(Ctx& c, int d, double now) {
  const EngineDesc& S = *c.S;
  while (c.free_gpus(d) > 0) {
    double size, enq;
    float netlat;
    int jid, ing;
    int jt;
    if (S.inf_priority && queue_pop(c, d, 0, size, netlat, jid, ing, enq)) jt = 0;
    else if (queue_pop(c, d, 1, size, netlat, jid, ing, enq)) jt = 1;
    else break;
    int n; double f;
    if (ALGO == A_CARBON_COST) {
      // the reference's drain path always uses the CARBON objective for
      // carbon_cost (simulator_paper_multi.py:909-920), unlike its admission
      // path which prefers cost when a price is set (:622-645)
      GridPick g = S.fp32_score
          ? wave_grid_argmin_f32(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                                 S.n_freq, S.max_gpj, 1, S.carbon[d], 0.0,
                                 false, 0.0)
          : wave_grid_argmin(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                             S.n_freq, S.max_gpj, 1, S.carbon[d], 0.0,
                             false, 0.0);
      n = g.n; f = g.f;
    } else if (ALGO == A_JOINT_NF || ALGO == A_BANDIT) {
      // explicit drain branches in the reference (:892-907); everything else
      // drains through the heuristic allocator (:922-927) — including debug
      decide_nf<ALGO>(c, d, jt, size, now, n, f);
    } else {
      n = heuristic_alloc(c, d, jt);
      f = c.hs->cur_freq[d];
    }
    n = max(1, min(n, c.free_gpus(d)));
    start_job(c, d, jt, size, netlat, jid, ing, n, f, now);
    if (c.lane == 0) c.hs->sum_wait += fmax(0.0, now - enq);
  }
}

// accrue energy + util across DCs to time t (lanes 0..n_dc-1)
__device__ void accrue_to(Ctx& c, double t) {
  const EngineDesc& S = *c.S;
  if (c.lane < S.n_dc) {
    double last = c.now;
    if (last < 0.0) {
      c.hs->util_begin[c.lane] = t;
    } else {
      double dt = fmax(0.0, t - last);
      c.hs->util_time[c.lane] += c.hs->busy[c.lane] * dt;
      c.hs->energy[c.lane] += c.dc_power(c.lane) * dt;
    }
  }
  lds_fence();
}

// emit one cluster-log row set (logging replica only; wave-cooperative counts)
__device__ void emit_cluster_rows(Ctx& c, double now) {
  const EngineDesc& S = *c.S;
  if (c.r != S.log_replica) return;
  int64_t base = (int64_t)c.r * S.total_slots;
  for (int d = 0; d < S.n_dc; ++d) {
    int lo = S.slot_off[d], hi = S.slot_off[d + 1];
    int cnt_inf = 0;
    for (int k = lo + c.lane; k < hi; k += SUBW)
      if (S.s_gpus[base + k] != 0 && S.s_jtype[base + k] == 0) cnt_inf++;
    cnt_inf = wave_sum_i32(cnt_inf);
    int run_total = c.hs->n_running[d];
    if (c.lane == 0) {
      int idx = *S.cl_count;
      if (idx < S.cl_cap) {
        double* row = &S.cl_rows[(int64_t)idx * 15];
        double begin = c.hs->util_begin[d];
        double elapsed = fmax(1e-9, now - (begin >= 0 ? begin : now));
        row[0] = now;
        row[1] = d;
        row[2] = c.hs->cur_freq[d];
        row[3] = c.hs->busy[d];
        row[4] = S.total_gpus[d] - c.hs->busy[d];
        row[5] = run_total;
        row[6] = cnt_inf;
        row[7] = run_total - cnt_inf;
        row[8] = c.hs->q_len[d * 2 + 0];
        row[9] = c.hs->q_len[d * 2 + 1];
        row[10] = S.total_gpus[d] ? (double)c.hs->busy[d] / S.total_gpus[d] : 0.0;
        row[11] = S.total_gpus[d]
            ? c.hs->util_time[d] / (S.total_gpus[d] * elapsed) : 0.0;
        row[12] = c.hs->acc_unit[d];
        row[13] = c.dc_power(d);
        row[14] = c.hs->energy[d];
        *S.cl_count = idx + 1;
      } else {
        atomicOr(&S.err[c.r], ERR_LOG_OVF);
      }
    }
    store_fence();
  }
}

__device__ void emit_job_row(Ctx& c, int d, int jt, int jid, int ing,
                             double size, double fused, int n, float netlat,
                             double start, double finish, int pcount) {
  const EngineDesc& S = *c.S;
  if (c.r != S.log_replica) return;
  if (c.lane == 0) {
    int idx = *S.jl_count;
    if (idx < S.jl_cap) {
      double* row = &S.jl_rows[(int64_t)idx * 11];
      row[0] = jid; row[1] = ing; row[2] = jt; row[3] = size; row[4] = d;
      row[5] = fused; row[6] = n; row[7] = netlat; row[8] = start;
      row[9] = finish; row[10] = pcount;
      *S.jl_count = idx + 1;
    } else {
      atomicOr(&S.err[c.r], ERR_LOG_OVF);
    }
  }
  store_fence();
}

// cap_greedy controller step at log ticks — reference-exact sorted-snapshot
// semantics (freq_load_agg.py:44-80 + simulator_paper_multi.py:229-315):
//   outer pass: snapshot every running job's f; build the full DOWN-ladder
//   atom list from that snapshot (atom (slot, k) = step lv[k] -> lv[k-1] for
//   k <= i0, rho frozen at snapshot ladder values);
//   inner loop: apply atoms in ascending (rho, task-order, step-order) —
//   emulated as a selection walk so no materialized sort is needed; an atom
//   whose target f is not strictly below the job's LIVE f is skipped (a
//   cheaper deeper atom may already have jumped the job lower); each applied
//   atom reschedules the job with the reference's exact progress arithmetic
//   and re-estimates total power before the next deficit check.
// Task order = DC-major, then running_jobs insertion order (s_seq); step
// order = i0-k ascending — exactly the reference's stable rho sort.
__device__ __attribute__((noinline)) void cap_greedy_control(Ctx& c,
                                                             double now) {
  // noinline ON PURPOSE: this runs once per log tick (cold), but inlined
  // it inflates live ranges across the whole advance loop — the
  // cap_greedy kernel instantiation measured 233 M ev/s vs 426 M for the
  // identical trajectory under cap_uniform (round2_algo_sweep.jsonl).
  const EngineDesc& S = *c.S;
  // sorted ladder (atoms_for_task sorts freq_levels)
  double lv[MAX_FREQ];
  for (int i = 0; i < S.n_freq; ++i) lv[i] = S.freq_levels[i];
  for (int i = 1; i < S.n_freq; ++i) {
    double x = lv[i];
    int j = i - 1;
    while (j >= 0 && lv[j] > x) { lv[j + 1] = lv[j]; --j; }
    lv[j + 1] = x;
  }
  double f_min = lv[0];
  double totalP = 0;
  for (int d = 0; d < S.n_dc; ++d) totalP += c.dc_power(d);
  if (totalP <= S.power_cap - 5.0) return;  // hysteresis (reference :235-237)
  double deficit = fmax(0.0, totalP - S.power_cap);
  if (deficit <= 1e-6) return;
  int64_t base = (int64_t)c.r * S.total_slots;
  int guard = 10000;
  while (deficit > 1e-6 && guard-- > 0) {
    // ---- snapshot: freeze each cappable task's f for this pass ----
    int have = 0;
    for (int k = c.lane; k < S.total_slots; k += SUBW) {
      double sf = 0.0;
      if (c.l_fin[k] < D_INF) {
        double fu = S.s_rec[base + k].fused;
        if (fu > f_min + 1e-12) { sf = fu; have = 1; }
      }
      S.snap_f[base + k] = sf;
    }
    store_fence();
    if (wave_sum_i32(have) == 0) break;  // no tasks
    bool applied_any = false;
    bool exhausted = false;
    // frozen-order selection walk: strictly-after (last_rho, last_key)
    double last_rho = -D_INF;
    long long last_key = -1;
    while (deficit > 1e-6) {
      double best_rho = D_INF;
      long long best_key = LLONG_MAX;
      int best_slot = -1, best_k = -1;
      for (int k = c.lane; k < S.total_slots; k += SUBW) {
        double sf = S.snap_f[base + k];
        if (sf <= 0.0) continue;
        int i0 = 0;
        double bd = 1e300;
        for (int q = 0; q < S.n_freq; ++q) {
          double df = fabs(lv[q] - sf);
          if (df < bd) { bd = df; i0 = q; }
        }
        if (i0 == 0) continue;
        int d = S.slot_dc[k];
        int jt = S.s_jtype[base + k];
        int n = S.s_gpus[base + k];
        long long seq = S.s_rec[base + k].seq;
        double curV = 1.0 / d_unit_time(n, lv[i0], c.lc3(d, jt));
        double curP = d_job_power(n, lv[i0], c.pc3(d, jt));
        for (int kk = i0; kk >= 1; --kk) {
          double V2 = 1.0 / d_unit_time(n, lv[kk - 1], c.lc3(d, jt));
          double P2 = d_job_power(n, lv[kk - 1], c.pc3(d, jt));
          double dV = fmax(0.0, curV - V2), dP = fmax(0.0, curP - P2);
          if (dV > 0 && dP >= 0) {
            double rho = dP / dV;
            long long key = ((long long)d << 40) | (seq << 8) | (i0 - kk);
            bool after = rho > last_rho || (rho == last_rho && key > last_key);
            if (after && (rho < best_rho ||
                          (rho == best_rho && key < best_key))) {
              best_rho = rho; best_key = key; best_slot = k; best_k = kk;
            }
          }
          curV = V2;
          curP = P2;
        }
      }
      // subgroup reduction on (rho, key)
#pragma unroll
      for (int off = SUBW / 2; off > 0; off >>= 1) {
        double orho = __shfl_xor(best_rho, off, 64);
        long long okey = __shfl_xor(best_key, off, 64);
        int oslot = __shfl_xor(best_slot, off, 64);
        int ok = __shfl_xor(best_k, off, 64);
        if (orho < best_rho || (orho == best_rho && okey < best_key)) {
          best_rho = orho; best_key = okey; best_slot = oslot; best_k = ok;
        }
      }
      if (best_rho >= D_INF) { exhausted = true; break; }  // atom list done
      last_rho = best_rho;
      last_key = best_key;
      // skip unless the atom still goes strictly DOWN from the live f
      SRec* bs = &S.s_rec[base + best_slot];
      double cur_f = bs->fused;
      double f_to = lv[best_k - 1];
      if (f_to >= cur_f - 1e-12) continue;
      // ---- apply: the reference's exact reschedule arithmetic ----
      int d = S.slot_dc[best_slot];
      int jt = S.s_jtype[base + best_slot];
      int n = S.s_gpus[base + best_slot];
      double T_old = d_unit_time(n, cur_f, c.lc3(d, jt));
      double rate_old = 1.0 / fmax(T_old, 1e-9);
      double size = bs->size;
      double dt = fmax(0.0, now - bs->lastupd);
      double done = fmin(size, bs->done + rate_old * dt);
      double units_left = fmax(0.0, size - done);
      double T_new = d_unit_time(n, f_to, c.lc3(d, jt));
      double rate_new = 1.0 / fmax(T_new, 1e-9);
      double finish_new = now + units_left / fmax(rate_new, 1e-9);
      if (c.lane == 0) {
        bs->done = done;
        bs->lastupd = now;
        bs->fused = f_to;
        c.l_fin[best_slot] = finish_new;
        c.hs->p_active[d] += d_job_power(n, f_to, c.pc3(d, jt)) -
                             d_job_power(n, cur_f, c.pc3(d, jt));
        c.hs->sum_tpt[d] += rate_new - rate_old;
      }
      store_fence();
      rescan_dc_min(c, d);
      applied_any = true;
      // exact total-power re-estimate after each applied atom (:300-307)
      totalP = 0;
      for (int dd = 0; dd < S.n_dc; ++dd) totalP += c.dc_power(dd);
      deficit = fmax(0.0, totalP - S.power_cap);
    }
    if (!applied_any) break;
    if (exhausted && deficit > 1e-6) continue;  // rebuild snapshot, next pass
  }
}

// ---------------- CHSAC-AF (RL) device helpers ----------------
// obs vector [now] + per-DC [total, busy, free, current_f, q_inf, q_train]
// (reference _upgr_obs, simulator_paper_multi.py:1041-1053); lanes 0..n_dc-1
// write their DC's 6 features in parallel.
__device__ void rl_build_obs(Ctx& c, double now, float* out) {
  const EngineDesc& S = *c.S;
  if (c.lane == 0) out[0] = (float)now;
  if (c.lane < S.n_dc) {
    float total = (float)S.total_gpus[c.lane];
    float busy = (float)c.hs->busy[c.lane];
    out[1 + 6 * c.lane + 0] = total;
    out[1 + 6 * c.lane + 1] = busy;
    out[1 + 6 * c.lane + 2] = fmaxf(0.0f, total - busy);
    out[1 + 6 * c.lane + 3] = c.hs->cur_freq[c.lane];
    out[1 + 6 * c.lane + 4] = (float)c.hs->q_len[c.lane * 2 + 0];
    out[1 + 6 * c.lane + 5] = (float)c.hs->q_len[c.lane * 2 + 1];
  }
  store_fence();
}

// approximate p99 (ms) from the per-replica log-binned latency histogram.
// The reference computes an exact percentile over a 2048-sample sliding
// window (:728-737); the histogram form is the batched approximation
// (documented divergence; exact at the oracle).
__device__ double rl_p99_ms(Ctx& c, int jt) {
  const EngineDesc& S = *c.S;
  long long total = S.lat_count[c.r * 2 + jt];
  if (S.exact_p99) {
    // np.percentile(buf, 99) over the sorted window: linear interpolation
    // at virtual index 0.99*(n-1)
    long long n = total < S.p99_win ? total : S.p99_win;
    if (n < 5) return -1.0;
    const double* a = &S.p99_sorted[(int64_t)(c.r * 2 + jt) * S.p99_win];
    double vi = 0.99 * (double)(n - 1);
    long long lo = (long long)vi;
    double frac = vi - (double)lo;
    double v = a[lo];
    if (lo + 1 < n) v = v + (a[lo + 1] - v) * frac;
    return v * 1000.0;
  }
  if (total < 5) return -1.0;
  long long target = (long long)(0.99 * (double)total);
  long long cum = 0;
  const int* h = &S.lat_hist[(c.r * 2 + jt) * LAT_BINS];
  for (int b = 0; b < LAT_BINS; ++b) {
    cum += h[b];
    if (cum > target) {
      // upper edge of bin b: 10^(-4 + 8*(b+1)/LAT_BINS) seconds
      double s = pow(10.0, -4.0 + 8.0 * (b + 1) / LAT_BINS);
      return s * 1000.0;
    }
  }
  return 1e7;
}

__device__ void rl_record_latency(Ctx& c, int jt, double sojourn_s) {
  const EngineDesc& S = *c.S;
  if (c.lane != 0) return;
  double l = log10(fmax(sojourn_s, 1e-9));
  int b = (int)((l + 4.0) / 8.0 * LAT_BINS);
  b = max(0, min(LAT_BINS - 1, b));
  S.lat_hist[(c.r * 2 + jt) * LAT_BINS + b] += 1;
  long long cnt = S.lat_count[c.r * 2 + jt];
  S.lat_count[c.r * 2 + jt] = cnt + 1;
  S.lat_sum[c.r * 2 + jt] += sojourn_s;
  if (S.exact_p99) {
    // maintain the sorted window (serial on lane 0 — parity-mode only)
    const int W = S.p99_win;
    double* srt = &S.p99_sorted[(int64_t)(c.r * 2 + jt) * W];
    double* ring = &S.p99_ring[(int64_t)(c.r * 2 + jt) * W];
    long long n = cnt < W ? cnt : W;
    if (cnt >= W) {
      // evict the oldest: find its slot in the sorted array, close the gap
      double old = ring[cnt % W];
      int p = 0;
      while (p < n && srt[p] < old) ++p;   // first slot holding `old`
      for (int q = p; q + 1 < n; ++q) srt[q] = srt[q + 1];
      n -= 1;
    }
    int p = 0;
    while (p < n && srt[p] < sojourn_s) ++p;
    for (int q = (int)n; q > p; --q) srt[q] = srt[q - 1];
    srt[p] = sojourn_s;
    ring[cnt % W] = sojourn_s;
  }
}

// masks (reference _upgr_masks :1055-1082): DC valid if it has free GPUs;
// g in 1..N valid if <= max free anywhere; g capped to 1 when recent p99 is
// comfortably under the SLA (training window preferred).
__device__ void rl_build_masks(Ctx& c, int& mdc, int& mg) {
  const EngineDesc& S = *c.S;
  int m1 = 0, max_free = 0;
  for (int d = 0; d < S.n_dc; ++d) {
    int free = c.free_gpus(d);
    if (free > 0) m1 |= (1 << d);
    max_free = max(max_free, free);
  }
  int m2 = 0;
  for (int g = 1; g <= S.max_gpj; ++g)
    if (g <= max_free) m2 |= (1 << (g - 1));
  int jt_buf = S.lat_count[c.r * 2 + 1] > 0 ? 1 : 0;
  double p99 = rl_p99_ms(c, jt_buf);
  if (p99 >= 0.0 && p99 < 0.9 * S.sla_p99_ms) {
    m2 = 0;
    if (1 <= max_free) m2 = 1;  // cap at a single GPU
  }
  mdc = m1;
  mg = m2;
}

// write an action request and stash the paused-event context
__device__ void rl_request(Ctx& c, int kind, double now, int jt, int ing,
                           double size, float netlat, int jid, int src_dc,
                           int from_inf, double enq) {
  const EngineDesc& S = *c.S;
  rl_build_obs(c, now, &S.req_obs[(int64_t)c.r * S.obs_dim]);
  int mdc, mg;
  rl_build_masks(c, mdc, mg);
  if (c.lane == 0) {
    S.req_mdc[c.r] = mdc;
    S.req_mg[c.r] = mg;
    S.pend_kind[c.r] = kind;
    S.pend_size[c.r] = size;
    S.pend_netlat[c.r] = netlat;
    S.pend_jid[c.r] = jid;
    S.pend_ing[c.r] = ing;
    S.pend_jt[c.r] = jt;
    S.pend_dc[c.r] = src_dc;
    S.pend_from_inf[c.r] = from_inf;
    S.pend_enq[c.r] = enq;
    S.req_flag[c.r] = REQ_PENDING;
  }
  store_fence();
}

// min n meeting the SLA at fixed f (reference _min_n_for_sla :1091-1096)
__device__ int rl_min_n_for_sla(Ctx& c, int d, int jt, double size, double f) {
  const EngineDesc& S = *c.S;
  for (int n = 1; n <= S.max_gpj; ++n)
    if (size * d_unit_time(n, f, c.lc3(d, jt)) * 1000.0 <= S.sla_p99_ms)
      return n;
  return S.max_gpj;
}

// energy-optimal f at n over the sorted ladder (deadline guard is vacuous:
// reference jobs never carry deadlines — Job.deadline is always None)
__device__ double rl_energy_freq(Ctx& c, int d, int jt, int n) {
  const EngineDesc& S = *c.S;
  return wave_energy_freq(c.pc3(d, jt), c.lc3(d, jt), S.freq_levels,
                          S.n_freq, n);
}

// ---------------- device-side actor forward + sampling ----------------
// One wave evaluates the full CHSAC actor for ONE observation:
// enc (obs->hid->hid->hid, ReLU) then the two categorical heads
// (hid->hid->n_dc / n_g).  Weights are [in][out]-major, so at every input
// index the SUBW lanes read consecutive floats (fully coalesced, L2-resident
// at ~1.1 MB total); activations live in LDS.  ~280k FMA per call.
constexpr int RL_MAX_HID = 256;
constexpr int RL_MAX_OBS = 64;
constexpr int RL_MAX_LOG = 16;

__device__ void rl_dense(const float* x, int in, const float* Wt,
                         const float* b, int out, float* y, int lane,
                         bool relu) {
  for (int j = lane; j < out; j += SUBW) {
    float acc = 0.0f;
    for (int i = 0; i < in; ++i) acc = fmaf(x[i], Wt[i * out + j], acc);
    acc += b[j];
    y[j] = relu ? fmaxf(acc, 0.0f) : acc;
  }
  lds_fence();
}

// logits for both heads; obs in LDS, dc logits at L[0..n_dc), g at L[8..)
__device__ void rl_logits_impl(const float* pw, int D, int H, int n_dc,
                               int n_g, const float* obs, float* A, float* B,
                               float* L, int lane) {
  const float *w1 = pw, *b1 = w1 + D * H;
  const float *w2 = b1 + H, *b2 = w2 + H * H;
  const float *w3 = b2 + H, *b3 = w3 + H * H;
  const float *hdc1 = b3 + H, *hdc1b = hdc1 + H * H;
  const float *hdc2 = hdc1b + H, *hdc2b = hdc2 + H * n_dc;
  const float *hg1 = hdc2b + n_dc, *hg1b = hg1 + H * H;
  const float *hg2 = hg1b + H, *hg2b = hg2 + H * n_g;
  rl_dense(obs, D, w1, b1, H, A, lane, true);
  rl_dense(A, H, w2, b2, H, B, lane, true);
  rl_dense(B, H, w3, b3, H, A, lane, true);        // h3 stays in A
  rl_dense(A, H, hdc1, hdc1b, H, B, lane, true);
  rl_dense(B, H, hdc2, hdc2b, n_dc, L, lane, false);
  rl_dense(A, H, hg1, hg1b, H, B, lane, true);
  rl_dense(B, H, hg2, hg2b, n_g, L + 8, lane, false);
}

__device__ void rl_actor_logits(Ctx& c) {
  const EngineDesc& S = *c.S;
  rl_logits_impl(S.pw, S.obs_dim, S.hid, S.n_dc, S.n_g, c.l_obs,
                 c.l_act_a, c.l_act_b, c.l_logits, c.lane);
}

// direct (key, ctr) uniform — distinct per-category Gumbel draws without
// advancing the per-replica stream n times serially
__device__ __forceinline__ double u01_at(uint64_t key, uint64_t ctr) {
  uint32_t w[4];
  philox4x32(key, ctr, w);
  return ((w[0] >> 5) * 67108864.0 + (w[1] >> 6)) *
         (1.0 / 9007199254740992.0);
}

// masked Gumbel-max categorical pick (equivalent in distribution to the host
// path's log_softmax+Gumbel argmax: the log-softmax shift cancels in the
// argmax).  mask==0 falls back to all-valid, mirroring the host guard.
// Wave-uniform: every lane computes the same winner from the same draws.
__device__ int rl_pick(Ctx& c, const float* logits, int mask, int n) {
  const EngineDesc& S = *c.S;
  if (mask == 0) mask = (1 << n) - 1;
  double best = -D_INF;
  int arg = 0;
  if (S.rl_det) {
    for (int j = 0; j < n; ++j) {
      double z = logits[j];
      if ((mask >> j & 1) && z > best) { best = z; arg = j; }
    }
    return arg;
  }
  uint64_t ctr0 = c.rng.ctr;
  c.rng.ctr += n;
  for (int j = 0; j < n; ++j) {
    if (!(mask >> j & 1)) continue;
    double u = fmax(u01_at(c.rng.key, ctr0 + j), 1e-20);
    double z = (double)logits[j] - log(-log(u));
    if (z > best) { best = z; arg = j; }
  }
  return arg;
}

// build obs+masks into LDS, run the actor, sample (a_dc, a_g)
__device__ void rl_serve_inline(Ctx& c, double now, int& a_dc, int& a_g,
                                int& mdc, int& mg) {
  const EngineDesc& S = *c.S;
  rl_build_obs(c, now, c.l_obs);
  rl_build_masks(c, mdc, mg);
  rl_actor_logits(c);
  a_dc = rl_pick(c, c.l_logits, mdc, S.n_dc);
  a_g = rl_pick(c, c.l_logits + 8, mg, S.n_g);
}

// start a job carrying an RL trace; returns chosen slot via start_job's path.
// (duplicates start_job, then fills the rl trace of the slot just used)
__device__ void rl_start_job(Ctx& c, int d, int jt, double size, float netlat,
                             int jid, int ing, int n, double f, double now,
                             const float* s0, int a_dc, int a_g,
                             int mdc, int mg, int n_rew,
                             double units_done = 0.0, int pcount = 0,
                             int has_rl = 1, double orig_start = -1.0) {
  const EngineDesc& S = *c.S;
  int64_t base = (int64_t)c.r * S.total_slots;
  int lo = S.slot_off[d], hi = S.slot_off[d + 1];
  int cand = INT_MAX;
  for (int k = lo + c.lane; k < hi; k += SUBW) {
    if (c.l_fin[k] >= D_INF) { cand = k; break; }
  }
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1)
    cand = min(cand, __shfl_xor(cand, off, 64));
  if (cand == INT_MAX) {
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_SLOT_OVF);
    return;
  }
  double T = d_unit_time(n, f, c.lc3(d, jt));
  // fresh starts: finish = now + size*T (reference _start_job_with_nf);
  // resumes (orig_start >= 0): the reference's exact arithmetic
  // units_left / max(1/T, 1e-9) and the ORIGINAL start time preserved
  // (reference _resume_preempted_job :362-387 — start_time not reset)
  double finish;
  if (orig_start >= 0.0) {
    double units_left = fmax(0.0, size - units_done);
    finish = now + units_left / fmax(1.0 / T, 1e-9);
  } else {
    finish = now + size * T;
  }
  // copy s0 trace (lane-parallel over obs_dim)
  for (int k = c.lane; k < S.obs_dim; k += SUBW)
    S.slot_s0[(base + cand) * S.obs_dim + k] = s0[k];
  if (c.lane == 0) {
    c.l_fin[cand] = finish;
    SRec* sr = &S.s_rec[base + cand];
    sr->start = orig_start >= 0.0 ? orig_start : now;
    sr->lastupd = now;
    sr->size = size;
    sr->fused = f;
    sr->netlat = netlat;
    sr->jid = jid;
    sr->done = units_done;
    sr->pcount = (unsigned char)pcount;
    sr->ing = (unsigned char)ing;
    sr->seq = ++c.hs->seq_ctr;
    sr->adc = (unsigned char)a_dc;
    sr->ag = (unsigned char)a_g;
    sr->mdc = (unsigned char)mdc;
    sr->mg = (unsigned char)mg;
    sr->has_rl = (unsigned char)has_rl;
    sr->nrew = (unsigned char)max(1, n_rew);
    S.s_gpus[base + cand] = (short)n;
    S.s_jtype[base + cand] = (char)jt;
    c.hs->busy[d] += n;
    c.hs->n_running[d] += 1;
    c.hs->p_active[d] += d_job_power(n, f, c.pc3(d, jt));
    c.hs->sum_tpt[d] += 1.0 / T;
    if (finish < c.hs->dc_minf[d]) {
      c.hs->dc_minf[d] = finish;
      c.hs->dc_mins[d] = cand;
    }
  }
  store_fence();
}

// emit a CHSAC transition (s0, s1, a, r, costs, masks) into the global ring
__device__ void rl_emit_transition(Ctx& c, const float* s0, int a_dc, int a_g,
                                   float r, float c_lat, float c_pow,
                                   float c_over, int mdc, int mg, double now) {
  const EngineDesc& S = *c.S;
  // s1 = obs at completion time; build into a scratch row first (reuse the
  // replica's request-obs row as scratch — safe: no pending request coexists
  // with a finish emission in the same event)
  float* s1 = &S.req_obs[(int64_t)c.r * S.obs_dim];
  rl_build_obs(c, now, s1);
  int idx = 0;
  if (c.lane == 0) idx = atomicAdd(S.tr_count, 1);
  // broadcast from THIS subgroup's base lane (absolute in-wave index)
  idx = __shfl(idx, (threadIdx.x & 63) & ~(SUBW - 1), 64);
  if (idx >= S.tr_cap) {
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_LOG_OVF);
    return;
  }
  for (int k = c.lane; k < S.obs_dim; k += SUBW) {
    S.tr_s0[(int64_t)idx * S.obs_dim + k] = s0[k];
    S.tr_s1[(int64_t)idx * S.obs_dim + k] = s1[k];
  }
  if (c.lane == 0) {
    S.tr_adc[idx] = (unsigned char)a_dc;
    S.tr_ag[idx] = (unsigned char)a_g;
    S.tr_r[idx] = r;
    S.tr_costs[idx * 3 + 0] = c_lat;
    S.tr_costs[idx * 3 + 1] = c_pow;
    S.tr_costs[idx * 3 + 2] = c_over;
    S.tr_mdc[idx] = (unsigned char)mdc;
    S.tr_mg[idx] = (unsigned char)mg;
  }
  store_fence();
}

// ---- elastic scaling (chsac): preempt all training jobs of a DC into the
// replica's pool, then reallocate them one by one with fresh RL actions
// (reference _preempt_all_training_jobs :396-409 + _rl_reallocate_training_jobs
// :498-534; a failed resume re-queues instead of stranding — oracle fix) ----
__device__ __attribute__((noinline)) int rl_elastic_preempt_all(
    Ctx& c, int d, double now) {
  const EngineDesc& S = *c.S;
  int64_t base = (int64_t)c.r * S.total_slots;
  int lo = S.slot_off[d], hi = S.slot_off[d + 1];
  int count = 0;
  // preempt in running_jobs INSERTION order (ascending s_seq), matching the
  // reference's dict walk; pool order drives the reallocation sequence
  for (;;) {   // uniform selection loop (rare event)
    int k = -1, best_seq = INT_MAX;
    for (int q = lo; q < hi; ++q) {
      if (S.s_gpus[base + q] == 0 || S.s_jtype[base + q] != 1) continue;
      int sq = S.s_rec[base + q].seq;
      if (sq < best_seq) { best_seq = sq; k = q; }
    }
    if (k < 0) break;
    if (count >= S.pp_cap) {
      if (c.lane == 0) atomicOr(&S.err[c.r], ERR_SLOT_OVF);
      break;
    }
    int n = S.s_gpus[base + k];
    SRec* sk = &S.s_rec[base + k];
    double f = sk->fused;
    double T = d_unit_time(n, f, c.lc3(d, 1));
    double done = sk->done +
                  fmax(0.0, now - sk->lastupd) / fmax(T, 1e-300);
    double size = sk->size;
    done = fmin(size, done);
    int64_t pb = (int64_t)c.r * S.pp_cap + count;
    if (c.lane == 0) {
      S.pp_size[pb] = size;
      S.pp_done[pb] = done;
      S.pp_start[pb] = sk->start;
      S.pp_netlat[pb] = sk->netlat;
      S.pp_jid[pb] = sk->jid;
      S.pp_ing[pb] = sk->ing;
      S.pp_dc[pb] = (unsigned char)d;
      S.pp_pcount[pb] = (unsigned char)(sk->pcount + 1);
      S.pp_adc[pb] = sk->adc;
      S.pp_ag[pb] = sk->ag;
      S.pp_nrew[pb] = sk->nrew;
      S.pp_has_rl[pb] = sk->has_rl;
      // free the slot + caches
      S.s_gpus[base + k] = 0;
      c.l_fin[k] = D_INF;
      c.hs->busy[d] -= n;
      c.hs->n_running[d] -= 1;
      c.hs->p_active[d] -= d_job_power(n, f, c.pc3(d, 1));
      c.hs->sum_tpt[d] -= 1.0 / T;
    }
    store_fence();
    // carry the RL trace (lane-parallel)
    for (int q = c.lane; q < S.obs_dim; q += SUBW)
      S.pp_s0[pb * S.obs_dim + q] = S.slot_s0[(base + k) * S.obs_dim + q];
    store_fence();
    ++count;
  }
  if (c.lane == 0) {
    S.pp_count[c.r] = count;
    S.pp_cursor[c.r] = 0;
  }
  store_fence();
  rescan_dc_min(c, d);
  return count;
}

// chsac drains AT MOST ONE queued job per finish via a fresh policy action
// (reference :849-890); returns true if a request was issued (pause)
__device__ bool rl_try_drain_request(Ctx& c, int d, double now) {
  const EngineDesc& S = *c.S;
  double qsize, qenq;
  float qnetlat;
  int qjid, qing;
  bool popped = false;
  int from_inf = 0;
  if (c.free_gpus(d) > 0) {
    if (S.inf_priority && queue_pop(c, d, 0, qsize, qnetlat, qjid, qing, qenq)) {
      popped = true;
      from_inf = 1;
    } else if (queue_pop(c, d, 1, qsize, qnetlat, qjid, qing, qenq)) {
      popped = true;
    }
  }
  if (popped)
    rl_request(c, PEND_DRAIN, now, from_inf ? 0 : 1, qing, qsize, qnetlat,
               qjid, d, from_inf, qenq);
  return popped;
}

// request the RL action for the pool entry at the cursor
__device__ void rl_request_realloc(Ctx& c, double now) {
  const EngineDesc& S = *c.S;
  int cur = S.pp_cursor[c.r];
  int64_t pb = (int64_t)c.r * S.pp_cap + cur;
  rl_request(c, PEND_REALLOC, now, 1, S.pp_ing[pb], S.pp_size[pb],
             S.pp_netlat[pb], S.pp_jid[pb], S.pp_dc[pb], 0, now);
}

// ---- action EXECUTION bodies, shared by the host resume path and the
// in-kernel serving path ----

// complete an RL-routed arrival: push the WAN transfer carrying the trace
// (reference :570-588 with the stashed action)
__device__ void rl_do_arrival(Ctx& c, double now, int jt, int ing,
                              double size, int jid, int a_dc, int a_g,
                              const float* s0, int mdc, int mg) {
  const EngineDesc& S = *c.S;
  int d_sel = a_dc;
  int n_sel = a_g + 1;
  double lnet = S.wan_lat[ing * S.n_dc + d_sel];
  double bw = S.wan_bw[ing * S.n_dc + d_sel];
  double xfer = bw > 0.0 ? S.payload_gb[jt] / bw : 0.0;
  int cand = INT_MAX;
  for (int k = c.lane; k < S.tcap; k += SUBW) {
    if (c.l_xt[k] >= D_INF) { cand = k; break; }
  }
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1)
    cand = min(cand, __shfl_xor(cand, off, 64));
  if (cand == INT_MAX) {
    if (c.lane == 0) atomicOr(&S.err[c.r], ERR_XFER_OVF);
    return;
  }
  int64_t at = (int64_t)c.r * S.tcap + cand;
  for (int k = c.lane; k < S.obs_dim; k += SUBW)
    S.x_s0[at * S.obs_dim + k] = s0[k];
  if (c.lane == 0) {
    c.l_xt[cand] = now + lnet + xfer;
    XRec* xr = &S.x_rec[at];
    xr->size = size;
    xr->netlat = (float)lnet;
    xr->jid = jid;
    xr->dc = (unsigned char)d_sel;
    xr->jtype = (unsigned char)jt;
    xr->ing = (unsigned char)ing;
    xr->nsel = (short)n_sel;
    xr->adc = (unsigned char)a_dc;
    xr->ag = (unsigned char)a_g;
    xr->mdc = (unsigned char)mdc;
    xr->mg = (unsigned char)mg;
    xr->has_rl = 1;
  }
  store_fence();
}

// execute a drain action on the popped job: start on the RL's target DC, or
// push back at the FRONT of the source queue when the target has no free
// GPUs (reference :849-890)
__device__ void rl_do_drain_action(Ctx& c, double now, int src_d,
                                   int from_inf, int jt, int ing, double size,
                                   float netlat, int jid, double enq,
                                   int a_dc, int a_g, const float* s0,
                                   int mdc, int mg) {
  const EngineDesc& S = *c.S;
  int d_tgt = a_dc;
  if (c.free_gpus(d_tgt) <= 0) {
    queue_push_front(c, src_d, from_inf ? 0 : 1, size, netlat, jid, ing, enq);
  } else {
    int n_sel = max(1, min(min(a_g + 1, c.free_gpus(d_tgt)), S.max_gpj));
    double f = rl_energy_freq(c, d_tgt, jt, n_sel);
    rl_start_job(c, d_tgt, jt, size, netlat, jid, ing, n_sel, f, now,
                 s0, a_dc, a_g, mdc, mg, n_sel);
    if (c.lane == 0) c.hs->sum_wait += fmax(0.0, now - enq);
  }
}

// in-kernel one-job queue drain (serve_device): pop, serve, execute — the
// replica continues its event loop with no host round-trip
__device__ void rl_drain_inline(Ctx& c, int d, double now) {
  const EngineDesc& S = *c.S;
  double qsize, qenq;
  float qnetlat;
  int qjid, qing;
  int from_inf = 0;
  bool popped = false;
  if (c.free_gpus(d) > 0) {
    if (S.inf_priority && queue_pop(c, d, 0, qsize, qnetlat, qjid, qing, qenq)) {
      popped = true;
      from_inf = 1;
    } else if (queue_pop(c, d, 1, qsize, qnetlat, qjid, qing, qenq)) {
      popped = true;
    }
  }
  if (!popped) return;
  int a_dc, a_g, mdc, mg;
  rl_serve_inline(c, now, a_dc, a_g, mdc, mg);
  rl_do_drain_action(c, now, d, from_inf, from_inf ? 0 : 1, qing, qsize,
                     qnetlat, qjid, qenq, a_dc, a_g, c.l_obs, mdc, mg);
}

// in-kernel elastic reallocation chain (serve_device): fresh obs + actor
// forward per pool entry, all within this launch (reference
// _rl_reallocate_training_jobs :498-534; failed resumes re-queue)
__device__ __attribute__((noinline)) void rl_realloc_inline(Ctx& c,
                                                            double now) {
  const EngineDesc& S = *c.S;
  int cnt = S.pp_count[c.r];
  for (int cur = 0; cur < cnt; ++cur) {
    int a_dc, a_g, mdc, mg;
    rl_serve_inline(c, now, a_dc, a_g, mdc, mg);
    int64_t pb = (int64_t)c.r * S.pp_cap + cur;
    int d_src = S.pp_dc[pb];
    if (c.free_gpus(d_src) <= 0) {
      queue_push(c, d_src, 1, S.pp_size[pb], S.pp_netlat[pb],
                 S.pp_jid[pb], S.pp_ing[pb], now);
    } else {
      int n_rl = max(1, min(min(a_g + 1, c.free_gpus(d_src)), S.max_gpj));
      double f = rl_energy_freq(c, d_src, 1, n_rl);
      rl_start_job(c, d_src, 1, S.pp_size[pb], S.pp_netlat[pb],
                   S.pp_jid[pb], S.pp_ing[pb], n_rl, f, now,
                   &S.pp_s0[pb * S.obs_dim], S.pp_adc[pb], S.pp_ag[pb],
                   mdc, mg, S.pp_nrew[pb], S.pp_done[pb], S.pp_pcount[pb],
                   S.pp_has_rl[pb], S.pp_start[pb]);
    }
  }
  if (c.lane == 0) {
    S.pp_count[c.r] = 0;
    S.pp_cursor[c.r] = 0;
  }
  store_fence();
}

// ---------------- the advance kernel ----------------
template <int ALGO>
__global__ void __launch_bounds__(THREADS_PER_BLOCK)
advance_kernel(EngineDesc S, double t_target, long long max_ev) {
  // SUBW lanes form one replica slot (64: wave-per-replica; 8: eight
  // replicas per wavefront)
  int slot_id = (blockIdx.x * blockDim.x + threadIdx.x) / SUBW;
  int lane = threadIdx.x & (SUBW - 1);
  if (slot_id >= S.n_rep) return;

  Ctx c;
  c.S = &S;
  c.r = slot_id;
  c.lane = lane;
  if (S.done[c.r]) return;
  // CHSAC fast-path: a replica still waiting for its policy response must
  // not advance (and must not round-trip the hot state)
  if (ALGO == A_CHSAC && S.pend_kind[c.r] != PEND_NONE &&
      S.req_flag[c.r] != REQ_READY)
    return;
  c.now = S.now[c.r];
  c.rng.key = S.seed ^ (0x9E3779B97F4A7C15ull * (uint64_t)(S.rep_id_offset + c.r));
  c.rng.ctr = S.rng_ctr[c.r];

  const int NS = S.n_streams;
  int64_t sbase = (int64_t)c.r * S.total_slots;
  long long n_events = 0;
  bool paused = false;
  bool skip_loop = false;  // realloc chain pending: bypass the event loop

  // ---- carve the dynamic-LDS region ----
  // SUBW==64: per replica, Hot + s_finish mirror + x_time mirror.
  // SUBW==8: Hot only (the mirrors would exceed the LDS budget at 8
  // replicas/wave); l_fin/l_xt then alias the replica's global rows.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  {
    size_t hot_sz = (sizeof(Hot) + 15) & ~size_t(15);
    size_t fin_sz = SUBW == 64 ? (size_t)S.total_slots * sizeof(double) : 0;
    size_t xt_sz = SUBW == 64 ? (size_t)S.tcap * sizeof(double) : 0;
    size_t rl_sz = ALGO == A_CHSAC
        ? (RL_MAX_OBS + 2 * RL_MAX_HID + RL_MAX_LOG) * sizeof(float) : 0;
    size_t stride = hot_sz + fin_sz + xt_sz + rl_sz;
    char* base = smem + (threadIdx.x / SUBW) * stride;
    c.hs = reinterpret_cast<Hot*>(base);
    if (SUBW == 64) {
      c.l_fin = reinterpret_cast<double*>(base + hot_sz);
      c.l_xt = reinterpret_cast<double*>(base + hot_sz + fin_sz);
    } else {
      c.l_fin = S.s_finish + sbase;
      c.l_xt = S.x_time + (int64_t)c.r * S.tcap;
    }
    if (ALGO == A_CHSAC) {
      float* rb = reinterpret_cast<float*>(base + hot_sz + fin_sz + xt_sz);
      c.l_obs = rb;
      c.l_act_a = rb + RL_MAX_OBS;
      c.l_act_b = c.l_act_a + RL_MAX_HID;
      c.l_logits = c.l_act_b + RL_MAX_HID;
    }
  }
  {
    Hot* h = c.hs;
    int nd = S.n_dc;
    if (SUBW == 64) {
      // slot finish times + transfer times into LDS (lane-strided)
      for (int k = lane; k < S.total_slots; k += SUBW)
        c.l_fin[k] = S.s_finish[sbase + k];
      for (int k = lane; k < S.tcap; k += SUBW)
        c.l_xt[k] = S.x_time[(int64_t)c.r * S.tcap + k];
    }
    for (int k = lane; k < NS; k += SUBW)
      h->arr_next[k] = S.arr_next[(int64_t)c.r * NS + k];
    if (lane < nd) {
      int rd = c.r * nd + lane;
      h->dc_minf[lane] = S.dc_min_finish[rd];
      h->dc_mins[lane] = S.dc_min_slot[rd];
      h->p_active[lane] = S.p_active[rd];
      h->sum_tpt[lane] = S.sum_tpt[rd];
      h->energy[lane] = S.energy_j[rd];
      h->util_time[lane] = S.util_time[rd];
      h->acc_unit[lane] = S.acc_unit[rd];
      h->util_begin[lane] = S.util_begin[rd];
      h->cur_freq[lane] = S.cur_freq[rd];
      h->busy[lane] = S.busy[rd];
      h->n_running[lane] = S.n_running[rd];
    }
    for (int k = lane; k < nd * 2; k += SUBW) {
      h->q_len[k] = S.q_len[c.r * nd * 2 + k];
      h->q_head[k] = S.q_head[c.r * nd * 2 + k];
    }
    if (lane == 0) {
      h->next_log = S.next_log[c.r];
      h->sum_lat = S.sum_lat[c.r];
      h->sum_lat_inf = S.sum_lat_inf[c.r];
      h->sum_wait = S.sum_wait[c.r];
      h->jobs_done = S.jobs_done[c.r];
      h->jobs_done_inf = S.jobs_done_inf[c.r];
      h->jid_ctr = S.jid_ctr[c.r];
      h->seq_ctr = S.seq_ctr[c.r];
    }
  }
  store_fence();  // global loads (vmcnt) AND the LDS writes (lgkmcnt)

  // ---- CHSAC: resume a paused action request ----
  if (ALGO == A_CHSAC && S.pend_kind[c.r] != PEND_NONE) {
    int a_dc = S.resp_dc[c.r];
    int a_g = S.resp_g[c.r];
    const float* s0 = &S.req_obs[(int64_t)c.r * S.obs_dim];
    int mdc = S.req_mdc[c.r], mg = S.req_mg[c.r];
    int pk = S.pend_kind[c.r];
    int jt = S.pend_jt[c.r];
    int ing = S.pend_ing[c.r];
    double size = S.pend_size[c.r];
    float netlat = S.pend_netlat[c.r];
    int jid = S.pend_jid[c.r];
    if (pk == PEND_ARRIVAL) {
      // complete the arrival: RL chose (dc, g); push the WAN transfer
      rl_do_arrival(c, c.now, jt, ing, size, jid, a_dc, a_g, s0, mdc, mg);
    } else if (pk == PEND_DRAIN) {
      // one queued job, RL chose a target DC + g
      rl_do_drain_action(c, c.now, S.pend_dc[c.r], S.pend_from_inf[c.r], jt,
                         ing, size, netlat, jid, S.pend_enq[c.r],
                         a_dc, a_g, s0, mdc, mg);
    } else {  // PEND_REALLOC: resume the pool entry at the cursor on its DC
      int cur = S.pp_cursor[c.r];
      int64_t pb = (int64_t)c.r * S.pp_cap + cur;
      int d_src = S.pp_dc[pb];
      if (c.free_gpus(d_src) <= 0) {
        // no free GPUs: re-queue on the training queue (oracle's fix of
        // reference Appendix A.6 job-stranding)
        queue_push(c, d_src, 1, S.pp_size[pb], S.pp_netlat[pb],
                   S.pp_jid[pb], S.pp_ing[pb], c.now);
      } else {
        int n_rl = max(1, min(min(a_g + 1, c.free_gpus(d_src)), S.max_gpj));
        double f = rl_energy_freq(c, d_src, 1, n_rl);
        rl_start_job(c, d_src, 1, S.pp_size[pb], S.pp_netlat[pb],
                     S.pp_jid[pb], S.pp_ing[pb], n_rl, f, c.now,
                     &S.pp_s0[pb * S.obs_dim], S.pp_adc[pb], S.pp_ag[pb],
                     mdc, mg, S.pp_nrew[pb], S.pp_done[pb], S.pp_pcount[pb],
                     S.pp_has_rl[pb], S.pp_start[pb]);
      }
      int nxt = cur + 1;
      if (nxt < S.pp_count[c.r]) {
        if (lane == 0) S.pp_cursor[c.r] = nxt;
        store_fence();
        // request the NEXT pool entry and skip straight to the hot-state
        // writeback (n_events = max_ev empties the event loop; an early
        // return here would LOSE the LDS-resident bookkeeping updates)
        rl_request_realloc(c, c.now);
        skip_loop = true;
        paused = true;
      } else {
        if (lane == 0) {
          S.pp_count[c.r] = 0;
          S.pp_cursor[c.r] = 0;
        }
        store_fence();
        // reallocation finished -> the deferred one-job queue drain
        if (rl_try_drain_request(c, d_src, c.now)) {
          skip_loop = true;
          paused = true;
        }
      }
    }
    if (!paused && lane == 0) {
      S.pend_kind[c.r] = PEND_NONE;
      S.req_flag[c.r] = REQ_IDLE;
    }
    store_fence();
  }

  while (!skip_loop && n_events < max_ev) {
    // serve-device staleness bound: yield the launch once the transition
    // ring is full enough for a host train round (checked every 32 events;
    // the read races benignly with other replicas' atomicAdds)
    if (ALGO == A_CHSAC && S.serve_device && S.tr_limit > 0 &&
        (n_events & 31) == 0 && *S.tr_count >= S.tr_limit)
      break;
    // ---- 1. next event: wave argmin over candidate sources ----
    // per-lane candidate: value + kind/idx
    double v = D_INF;
    int kind = -1, idx = -1;
    // arrival streams (strided)
    for (int k = lane; k < NS; k += SUBW) {
      double t = c.hs->arr_next[k];
      if (t < v) { v = t; kind = 0; idx = k; }
    }
    // log tick (one candidate)
    if (lane == 0) {
      double t = c.hs->next_log;
      if (t < v) { v = t; kind = 3; idx = 0; }
    }
    // dc min finishes (strided; n_dc <= 8 <= SUBW so one pass)
    for (int k = lane; k < S.n_dc; k += SUBW) {
      double t = c.hs->dc_minf[k];
      if (t < v) { v = t; kind = 2; idx = c.hs->dc_mins[k]; }
    }
    // transfers (strided over the mirror)
    for (int k = lane; k < S.tcap; k += SUBW) {
      double t = c.l_xt[k];
      if (t < v) { v = t; kind = 1; idx = k; }
    }
    int wl;
    double t_min = wave_argmin_f64(v, wl);
    kind = __shfl(kind, wl, 64);
    idx = __shfl(idx, wl, 64);

    if (t_min > S.end_time) {
      // ---- end of simulation for this replica: final flush ----
      if (lane < S.n_dc) {
        double last = c.now;
        if (last >= 0.0 && last < S.end_time) {
          c.hs->util_time[lane] += c.hs->busy[lane] * (S.end_time - last);
          // reference quirk: the final accrue_energy(end) uses the BASELINE
          // idle/sleep + p_peak*f^alpha model (models.py:82-91)
          double f = c.hs->cur_freq[lane];
          int active = c.hs->busy[lane];
          int idlec = S.total_gpus[lane] - active;
          double pa = active * (S.p_idle[lane] +
                                S.p_peak[lane] * pow(f, S.pow_alpha[lane]));
          double pi = idlec * (S.power_gating[lane] ? S.p_sleep[lane] : S.p_idle[lane]);
          c.hs->energy[lane] += (pa + pi) * (S.end_time - last);
        }
      }
      if (lane == 0) S.done[c.r] = 1;
      store_fence();
      break;
    }
    if (t_min > t_target) break;  // chunk boundary

    // ---- 2. accrue energy + util to t_min ----
    accrue_to(c, t_min);
    c.now = t_min;
    n_events++;

    // ---- 3. dispatch ----
    if (kind == 0) {
      // ===== arrival at ingress stream idx =====
      int ing = idx >> 1;
      int jt = idx & 1;
      int jid = c.hs->jid_ctr + 1;
      if (lane == 0) c.hs->jid_ctr = jid;
      double size;
      int trace_d = -1;
      if (S.trace_mode) {
        // replay mode: this arrival's size, routed DC and the stream's next
        // time come from the recorded trace
        int64_t tb = ((int64_t)c.r * S.n_streams + idx) * S.trace_cap;
        int posn = S.trace_pos[(int64_t)c.r * S.n_streams + idx];
        size = S.trace_size[tb + posn];
        trace_d = S.trace_dc[tb + posn];
        int nxt = posn + 1;
        double t_next = nxt < S.trace_cap ? S.trace_time[tb + nxt] : D_INF;
        if (lane == 0) {
          S.trace_pos[(int64_t)c.r * S.n_streams + idx] = nxt;
          c.hs->arr_next[idx] = t_next;
        }
        lds_fence();
      } else {
        size = jt == 0 ? rpareto_inf(c.rng)
                       : fmax(0.1, rlognormal(c.rng, log(50000.0), 0.4));
      }
      if (ALGO == A_CHSAC) {
        // schedule the next arrival first (RNG order differs from the scalar
        // engines; distributionally identical), then pause for the policy
        double ia_rl = D_INF;
        if (!S.trace_mode) {
          double rate = S.arr_rate[jt];
          int mode = S.arr_mode[jt];
          double amp = S.arr_amp[jt];
          double period = S.arr_period[jt];
          if (mode == 0 && rate > 0) {
            ia_rl = rexp(c.rng, rate);
          } else if (mode == 1) {
            double max_rate = rate * (1.0 + fabs(amp));
            if (max_rate > 0) {
              for (int it = 0; it < 4096; ++it) {
                double w = rexp(c.rng, max_rate);
                double lam = fmax(0.0, rate * (1.0 + amp *
                    sin(2.0 * M_PI * fmod(t_min + w, period) / period)));
                if (u01(c.rng) <= lam / max_rate) { ia_rl = w; break; }
              }
            }
          }
        }
        if (lane == 0 && !S.trace_mode) c.hs->arr_next[idx] = t_min + ia_rl;
        lds_fence();
        if (S.serve_device) {
          // in-kernel policy: forward + sample + execute, no pause
          int a_dc, a_g, mdc, mg;
          rl_serve_inline(c, t_min, a_dc, a_g, mdc, mg);
          rl_do_arrival(c, t_min, jt, ing, size, jid, a_dc, a_g, c.l_obs,
                        mdc, mg);
        } else {
          rl_request(c, PEND_ARRIVAL, t_min, jt, ing, size, 0.0f, jid,
                     -1, 0, t_min);
          paused = true;
          break;
        }
      } else {
      // routing
      int d_sel;
      if (S.trace_mode && trace_d >= 0) {
        d_sel = trace_d;
      } else if (ALGO == A_ECO_ROUTE) {
        double price = S.eco_obj == 2 ? c.price_kwh(t_min) : 0.0;
        if (SUBW == 64) {
          // all 64 lanes score: lane = d*8 + (n-1) covers (DC, n); each lane
          // reduces over the frequency ladder, then an 8-lane subgroup min
          // per DC and a wave argmin pick the DC (first-minimum = lowest DC
          // index, matching the scalar scan order).
          double score = D_INF;
          {
            int d = lane >> 3;
            int n = (lane & 7) + 1;
            if (d < S.n_dc && n <= S.max_gpj) {
              const double* pcf = c.pc3(d, jt);
              const double* tcf = c.lc3(d, jt);
              double best = D_INF;
              for (int q = 0; q < S.n_freq; ++q) {
                double f = S.freq_levels[q];
                double T = d_unit_time(n, f, tcf);
                double E = d_job_power(n, f, pcf) * T;
                double sc;
                if (S.eco_obj == 1) sc = E * S.carbon[d];
                else if (S.eco_obj == 2) sc = (E / 3.6e6) * price;
                else sc = E;
                if (sc < best) best = sc;
              }
              score = best * size;  // relative order preserved per objective
            }
          }
#pragma unroll
          for (int off = 1; off < 8; off <<= 1)
            score = fmin(score, __shfl_xor(score, off, 64));
          if (lane & 7) score = D_INF;
          int dl;
          wave_argmin_f64(score, dl);
          d_sel = dl >> 3;
        } else {
          // SUBW==8: lane d scores DC d serially over the grid
          double score = D_INF;
          if (lane < S.n_dc) {
            int d = lane;
            const double* pcf = c.pc3(d, jt);
            const double* tcf = c.lc3(d, jt);
            double best = D_INF;
            for (int n = 1; n <= S.max_gpj; ++n)
              for (int q = 0; q < S.n_freq; ++q) {
                double f = S.freq_levels[q];
                double T = d_unit_time(n, f, tcf);
                double E = d_job_power(n, f, pcf) * T;
                double sc;
                if (S.eco_obj == 1) sc = E * S.carbon[d];
                else if (S.eco_obj == 2) sc = (E / 3.6e6) * price;
                else sc = E;
                if (sc < best) best = sc;
              }
            score = best * size;
          }
          int dl;
          wave_argmin_f64(score, dl);
          d_sel = dl & (SUBW - 1);
        }
      } else {
        d_sel = (int)rbelow(c.rng, (uint32_t)S.n_dc);
      }
      double lnet = S.wan_lat[ing * S.n_dc + d_sel];
      double bw = S.wan_bw[ing * S.n_dc + d_sel];
      double xfer = bw > 0.0 ? S.payload_gb[jt] / bw : 0.0;
      // push transfer record
      int cand = INT_MAX;
      for (int k = lane; k < S.tcap; k += SUBW) {
        if (c.l_xt[k] >= D_INF) { cand = k; break; }
      }
#pragma unroll
      for (int off = SUBW / 2; off > 0; off >>= 1)
        cand = min(cand, __shfl_xor(cand, off, 64));
      if (cand == INT_MAX) {
        if (lane == 0) atomicOr(&S.err[c.r], ERR_XFER_OVF);
      } else if (lane == 0) {
        int64_t at = (int64_t)c.r * S.tcap + cand;
        c.l_xt[cand] = t_min + lnet + xfer;
        XRec* xr = &S.x_rec[at];
        xr->size = size;
        xr->netlat = (float)lnet;
        xr->jid = jid;
        xr->dc = (unsigned char)d_sel;
        xr->jtype = (unsigned char)jt;
        xr->ing = (unsigned char)ing;
        xr->has_rl = 0;
      }
      store_fence();
      // next arrival for this stream (faithful non-accumulating thinning;
      // reference arrivals.py:35-48); in replay mode it was already set
      double ia = D_INF;
      if (!S.trace_mode) {
        double rate = S.arr_rate[jt];
        int mode = S.arr_mode[jt];
        double amp = S.arr_amp[jt];
        double period = S.arr_period[jt];
        if (mode == 0 && rate > 0) {
          ia = rexp(c.rng, rate);
        } else if (mode == 1) {
          double max_rate = rate * (1.0 + fabs(amp));
          if (max_rate > 0) {
            for (int it = 0; it < 4096; ++it) {
              double w = rexp(c.rng, max_rate);
              double lam = fmax(0.0, rate * (1.0 + amp *
                  sin(2.0 * M_PI * fmod(t_min + w, period) / period)));
              if (u01(c.rng) <= lam / max_rate) { ia = w; break; }
            }
          }
        }
      }
      if (lane == 0 && !S.trace_mode) c.hs->arr_next[idx] = t_min + ia;
      lds_fence();
      }  // end non-chsac arrival path

    } else if (kind == 1) {
      // ===== WAN transfer complete: admission =====
      int64_t at = (int64_t)c.r * S.tcap + idx;
      XRec xr = S.x_rec[at];
      int d = xr.dc;
      int jt = xr.jtype;
      double size = xr.size;
      float netlat = xr.netlat;
      int jid = xr.jid;
      int ing = xr.ing;
      if (lane == 0) c.l_xt[idx] = D_INF;
      lds_fence();
      if (c.free_gpus(d) > 0) {
        if (ALGO == A_CHSAC && xr.has_rl) {
          // RL-chosen n (clamped), energy-optimal f (reference :646-667)
          int n = max(1, min(min((int)xr.nsel, c.free_gpus(d)), S.max_gpj));
          double f = rl_energy_freq(c, d, jt, n);
          rl_start_job(c, d, jt, size, netlat, jid, ing, n, f, t_min,
                       &S.x_s0[at * S.obs_dim], xr.adc, xr.ag,
                       xr.mdc, xr.mg, (int)xr.nsel);
        } else {
          int n; double f;
          decide_nf<ALGO>(c, d, jt, size, t_min, n, f);
          n = max(1, min(n, c.free_gpus(d)));
          start_job(c, d, jt, size, netlat, jid, ing, n, f, t_min);
        }
      } else {
        // queueing drops the RL trace: a later chsac drain assigns a fresh
        // observation/action (reference :849-889 overwrites them)
        queue_push(c, d, jt, size, netlat, jid, ing, t_min);
      }

    } else if (kind == 2) {
      // ===== job finish =====
      int slot = idx;
      int d = S.slot_dc[slot];
      int64_t at = sbase + slot;
      int jt = S.s_jtype[at];
      int n = S.s_gpus[at];
      SRec srv = S.s_rec[at];
      double size = srv.size;
      double fused = srv.fused;
      double start = srv.start;
      float netlat = srv.netlat;
      int jid = srv.jid;
      int ingr = srv.ing;
      int pcount = srv.pcount;
      double T = d_unit_time(n, fused, c.lc3(d, jt));
      if (lane == 0) {
        c.l_fin[slot] = D_INF;
        S.s_gpus[at] = 0;
        c.hs->busy[d] = max(0, c.hs->busy[d] - n);
        c.hs->n_running[d] -= 1;
        c.hs->p_active[d] -= d_job_power(n, fused, c.pc3(d, jt));
        c.hs->sum_tpt[d] -= 1.0 / T;
        // remainder job-units: window = finish mod log_interval (quirk)
        c.hs->acc_unit[d] += (1.0 / T) * fmod(t_min, S.log_interval);
        // metrics
        c.hs->jobs_done += 1;
        c.hs->sum_lat += t_min - start;
        if (jt == 0) {
          c.hs->jobs_done_inf += 1;
          c.hs->sum_lat_inf += t_min - start;
        }
        if (ALGO == A_BANDIT) {
          // reward = -E_pred (energy per unit at the used f); match the arm
          // by NEAREST ladder frequency (f_used is stored f32, so exact f64
          // comparison would never hit)
          double E = d_job_power(n, fused, c.pc3(d, jt)) * T;
          int64_t ab = ((int64_t)c.r * S.n_dc + d) * 2 + jt;
          int bk = 0;
          double bd = 1e300;
          for (int k = 0; k < S.n_freq; ++k) {
            double diff = fabs(S.freq_levels[k] - fused);
            if (diff < bd) { bd = diff; bk = k; }
          }
          S.b_n[ab * S.n_freq + bk] += 1;
          S.b_s[ab * S.n_freq + bk] += -E;
        }
      }
      store_fence();
      emit_job_row(c, d, jt, jid, ingr, size, fused, n, netlat, start, t_min,
                   pcount);
      rescan_dc_min(c, d);
      if (ALGO == A_CHSAC) {
        // record latency, then build the transition (reference :718-800)
        double sojourn = fmax(0.0, t_min - start);
        rl_record_latency(c, jt, sojourn);
        store_fence();
        if (srv.has_rl) {
          double E_pred = d_job_power(n, fused, c.pc3(d, jt)) * T;  // J/unit
          double E_unit_kwh = (E_pred * (double)size / 3.6e6) /
                              ((double)size + 1e-9);
          int n_rew = max(1, (int)srv.nrew);
          float r = (float)(-E_unit_kwh + 0.05 * (1.0 / n_rew));
          double p99 = rl_p99_ms(c, jt);
          if (p99 < 0.0) p99 = sojourn * 1000.0;  // <5 samples fallback
          double P_now = c.dc_power(d);
          int n_min = rl_min_n_for_sla(c, d, jt, size, fused);
          float c_over = (float)max(0, n - n_min);
          int mdc, mg;
          rl_build_masks(c, mdc, mg);  // masks at completion (reference :793)
          rl_emit_transition(c, &S.slot_s0[at * S.obs_dim],
                             srv.adc, srv.ag, r,
                             (float)p99, (float)P_now, c_over, mdc, mg, t_min);
          if (lane == 0) S.s_rec[at].has_rl = 0;
          store_fence();
        }
        // elastic scaling: on a training completion with other training jobs
        // still running in this DC, preempt them all and reallocate via
        // fresh RL actions (reference :829-837; gated to chsac + flag)
        if (S.elastic && jt == 1) {
          int n_train = 0;
          for (int k = S.slot_off[d]; k < S.slot_off[d + 1]; ++k)
            if (S.s_gpus[sbase + k] != 0 && S.s_jtype[sbase + k] == 1)
              ++n_train;
          if (n_train > 1) {
            int cnt = rl_elastic_preempt_all(c, d, t_min);
            if (cnt > 0) {
              if (S.serve_device) {
                rl_realloc_inline(c, t_min);  // whole chain, this launch
              } else {
                rl_request_realloc(c, t_min);
                paused = true;
                break;
              }
            }
          }
        }
        if (S.serve_device) {
          rl_drain_inline(c, d, t_min);
        } else if (rl_try_drain_request(c, d, t_min)) {
          paused = true;
          break;
        }
      } else {
        drain_queues<ALGO>(c, d, t_min);
        rescan_dc_min(c, d);
      }

    } else {
      // ===== log tick =====
      // power-cap control first (reference :463-465 order)
      if (S.power_cap > 0) {
        if (ALGO == A_CAP_GREEDY) {
          cap_greedy_control(c, t_min);
        } else if (ALGO == A_ECO_ROUTE || ALGO == A_CARBON_COST) {
          if (lane < S.n_dc) {
            if (c.hs->busy[lane] == 0) {
              double fm = S.freq_levels[0];
              for (int k = 1; k < S.n_freq; ++k) fm = fmin(fm, S.freq_levels[k]);
              c.hs->cur_freq[lane] = fm;
            }
          }
          store_fence();
        }
        // cap_uniform: exact no-op (per-job f_used power model; see oracle)
      }
      // acc_job_unit for running jobs: cached sum_tpt * interval
      if (lane < S.n_dc) {
        c.hs->acc_unit[lane] += c.hs->sum_tpt[lane] * S.log_interval;
      }
      lds_fence();
      emit_cluster_rows(c, t_min);
      if (lane == 0) c.hs->next_log = t_min + S.log_interval;
      lds_fence();
    }
  }

  // ---- write the hot state back to HBM ----
  {
    Hot* h = c.hs;
    int nd = S.n_dc;
    if (SUBW == 64) {
      for (int k = lane; k < S.total_slots; k += SUBW)
        S.s_finish[sbase + k] = c.l_fin[k];
      for (int k = lane; k < S.tcap; k += SUBW)
        S.x_time[(int64_t)c.r * S.tcap + k] = c.l_xt[k];
    }
    for (int k = lane; k < NS; k += SUBW)
      S.arr_next[(int64_t)c.r * NS + k] = h->arr_next[k];
    if (lane < nd) {
      int rd = c.r * nd + lane;
      S.dc_min_finish[rd] = h->dc_minf[lane];
      S.dc_min_slot[rd] = h->dc_mins[lane];
      S.p_active[rd] = h->p_active[lane];
      S.sum_tpt[rd] = h->sum_tpt[lane];
      S.energy_j[rd] = h->energy[lane];
      S.util_time[rd] = h->util_time[lane];
      S.acc_unit[rd] = h->acc_unit[lane];
      S.util_begin[rd] = h->util_begin[lane];
      S.cur_freq[rd] = h->cur_freq[lane];
      S.busy[rd] = h->busy[lane];
      S.n_running[rd] = h->n_running[lane];
    }
    for (int k = lane; k < nd * 2; k += SUBW) {
      S.q_len[c.r * nd * 2 + k] = h->q_len[k];
      S.q_head[c.r * nd * 2 + k] = h->q_head[k];
    }
  }
  if (lane == 0) {
    S.next_log[c.r] = c.hs->next_log;
    S.now[c.r] = c.now;
    S.rng_ctr[c.r] = c.rng.ctr;
    S.ev_count[c.r] += n_events;
    S.sum_lat[c.r] = c.hs->sum_lat;
    S.sum_lat_inf[c.r] = c.hs->sum_lat_inf;
    S.sum_wait[c.r] = c.hs->sum_wait;
    S.jobs_done[c.r] = c.hs->jobs_done;
    S.jobs_done_inf[c.r] = c.hs->jobs_done_inf;
    S.jid_ctr[c.r] = c.hs->jid_ctr;
    S.seq_ctr[c.r] = c.hs->seq_ctr;
  }
}

// NEGATIVE RESULT (kept out): a grid-stride launcher over an extracted
// advance_one device function, with the chsac grid capped at resident
// capacity to free the workgroup dispatcher for concurrent train kernels.
// The restructure tanked the flagship 400 -> 247 M ev/s (65 k replicas:
// 551 -> 302 M) with OR without always_inline — the loop-wrapped body
// spills the wave-uniform scalars, the same failure mode as round 1's
// noinline experiment.  Train/sim concurrency remains bounded by
// dispatch behavior; the update-rate controller owns that tradeoff.

// ---------------- MFMA batched actor forward ----------------
// Matrix-core path for BATCHED policy evaluation (host-side serving /
// offline eval / act-batch): Y = relu(X W^T + b) chained through the whole
// actor on fp32 MFMA tiles (v_mfma_f32_16x16x4_f32 — exact f32, so results
// match the fmaf-chain serving path modulo accumulation order).  One
// workgroup of 16 waves owns 16 batch rows; each wave produces one 16-wide
// column tile per layer; activations ping-pong through padded LDS.
// Fragment maps (cdna4_isa.md §10): A[i=l&15][k=l>>4], B[k=l>>4][j=l&15],
// C/D col=l&15, row=(l>>4)*4+reg.
constexpr int MF_STRIDE = 257;  // 16-row LDS activation stride (bank-spread)
using mf_acc = __attribute__((ext_vector_type(4))) float;

__device__ void mfma_layer(const float* Wt, const float* b, int in, int out,
                           const float* X, float* Y, int wave, int lane,
                           bool relu, int rows) {
  int jt = wave * 16;
  if (jt < out) {
    mf_acc acc = {0.f, 0.f, 0.f, 0.f};
    int i = lane & 15, kw = lane >> 4, j = jt + (lane & 15);
    for (int k0 = 0; k0 < in; k0 += 4) {
      int k = k0 + kw;
      float a = (k < in) ? X[i * MF_STRIDE + k] : 0.0f;
      float w = (k < in && j < out) ? Wt[k * out + j] : 0.0f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, w, acc, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      if (j < out && row < rows) {
        float y = acc[r] + b[j];
        Y[row * MF_STRIDE + j] = relu ? fmaxf(y, 0.0f) : y;
      }
    }
  }
  __syncthreads();
}

__global__ void __launch_bounds__(1024)
rl_forward_mfma_kernel(const float* pw, const float* obs, int B, int obs_dim,
                       int hid, int n_dc, int n_g, float* out_dc,
                       float* out_g) {
  __shared__ float bufA[16 * MF_STRIDE];
  __shared__ float bufB[16 * MF_STRIDE];
  int wave = threadIdx.x / 64;
  int lane = threadIdx.x & 63;
  int row0 = blockIdx.x * 16;
  int rows = min(16, B - row0);
  if (rows <= 0) return;
  // stage the obs tile (zero-padded)
  for (int idx = threadIdx.x; idx < 16 * MF_STRIDE; idx += blockDim.x)
    bufA[idx] = 0.0f;
  __syncthreads();
  for (int idx = threadIdx.x; idx < rows * obs_dim; idx += blockDim.x) {
    int r = idx / obs_dim, k = idx % obs_dim;
    bufA[r * MF_STRIDE + k] = obs[(int64_t)(row0 + r) * obs_dim + k];
  }
  __syncthreads();
  const int H = hid, D = obs_dim;
  const float *w1 = pw, *b1 = w1 + D * H;
  const float *w2 = b1 + H, *b2 = w2 + H * H;
  const float *w3 = b2 + H, *b3 = w3 + H * H;
  const float *hdc1 = b3 + H, *hdc1b = hdc1 + H * H;
  const float *hdc2 = hdc1b + H, *hdc2b = hdc2 + H * n_dc;
  const float *hg1 = hdc2b + n_dc, *hg1b = hg1 + H * H;
  const float *hg2 = hg1b + H, *hg2b = hg2 + H * n_g;
  mfma_layer(w1, b1, D, H, bufA, bufB, wave, lane, true, rows);
  mfma_layer(w2, b2, H, H, bufB, bufA, wave, lane, true, rows);
  mfma_layer(w3, b3, H, H, bufA, bufB, wave, lane, true, rows);  // h3 in B
  mfma_layer(hdc1, hdc1b, H, H, bufB, bufA, wave, lane, true, rows);
  // dc logits land in the first n_dc columns of a fresh LDS row set; reuse
  // bufA's upper half is unsafe (still holding hdc1 output) — write the
  // small head outputs straight to global from the accumulator instead:
  {
    int jt = wave * 16;
    if (jt < n_dc) {
      mf_acc acc = {0.f, 0.f, 0.f, 0.f};
      int i = lane & 15, kw = lane >> 4, j = jt + (lane & 15);
      for (int k0 = 0; k0 < H; k0 += 4) {
        int k = k0 + kw;
        float a = bufA[i * MF_STRIDE + k];
        float w = (j < n_dc) ? hdc2[k * n_dc + j] : 0.0f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, w, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = (lane >> 4) * 4 + r;
        if (j < n_dc && row < rows)
          out_dc[(int64_t)(row0 + row) * n_dc + j] = acc[r] + hdc2b[j];
      }
    }
    __syncthreads();
  }
  mfma_layer(hg1, hg1b, H, H, bufB, bufA, wave, lane, true, rows);
  {
    int jt = wave * 16;
    if (jt < n_g) {
      mf_acc acc = {0.f, 0.f, 0.f, 0.f};
      int i = lane & 15, kw = lane >> 4, j = jt + (lane & 15);
      for (int k0 = 0; k0 < H; k0 += 4) {
        int k = k0 + kw;
        float a = bufA[i * MF_STRIDE + k];
        float w = (j < n_g) ? hg2[k * n_g + j] : 0.0f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, w, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = (lane >> 4) * 4 + r;
        if (j < n_g && row < rows)
          out_g[(int64_t)(row0 + row) * n_g + j] = acc[r] + hg2b[j];
      }
    }
  }
}

std::vector<torch::Tensor> rl_forward_mfma(torch::Tensor pw,
                                           torch::Tensor obs,
                                           int64_t hid, int64_t n_dc,
                                           int64_t n_g) {
  TORCH_CHECK(pw.is_cuda() && obs.is_cuda() && pw.dtype() == torch::kFloat32
              && obs.dtype() == torch::kFloat32 && obs.is_contiguous());
  TORCH_CHECK(hid % 16 == 0 && hid <= 256 && n_dc <= 16 && n_g <= 16);
  int B = obs.size(0), D = obs.size(1);
  auto out_dc = torch::empty({B, n_dc}, obs.options());
  auto out_g = torch::empty({B, n_g}, obs.options());
  int blocks = (B + 15) / 16;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rl_forward_mfma_kernel, dim3(blocks), dim3(1024), 0,
                     stream, pw.data_ptr<float>(), obs.data_ptr<float>(), B,
                     D, (int)hid, (int)n_dc, (int)n_g,
                     out_dc.data_ptr<float>(), out_g.data_ptr<float>());
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("rl_forward_mfma_kernel: ") +
                             hipGetErrorString(e));
  return {out_dc, out_g};
}

// standalone batched actor forward (test/verification path): one subwave
// slot per observation row, same device math as the in-engine serving
__global__ void __launch_bounds__(THREADS_PER_BLOCK)
rl_forward_kernel(const float* pw, const float* obs, int B, int obs_dim,
                  int hid, int n_dc, int n_g, float* out_dc, float* out_g) {
  int slot = (blockIdx.x * blockDim.x + threadIdx.x) / SUBW;
  int lane = threadIdx.x & (SUBW - 1);
  extern __shared__ __attribute__((aligned(16))) char smem[];
  size_t per = (RL_MAX_OBS + 2 * RL_MAX_HID + RL_MAX_LOG) * sizeof(float);
  float* rb = reinterpret_cast<float*>(smem + (threadIdx.x / SUBW) * per);
  if (slot >= B) return;
  float* l_obs = rb;
  float* A = rb + RL_MAX_OBS;
  float* Bv = A + RL_MAX_HID;
  float* L = Bv + RL_MAX_HID;
  for (int k = lane; k < obs_dim; k += SUBW)
    l_obs[k] = obs[(int64_t)slot * obs_dim + k];
  lds_fence();
  rl_logits_impl(pw, obs_dim, hid, n_dc, n_g, l_obs, A, Bv, L, lane);
  for (int k = lane; k < n_dc; k += SUBW)
    out_dc[(int64_t)slot * n_dc + k] = L[k];
  for (int k = lane; k < n_g; k += SUBW)
    out_g[(int64_t)slot * n_g + k] = L[8 + k];
}

std::vector<torch::Tensor> rl_forward_debug(torch::Tensor pw,
                                            torch::Tensor obs,
                                            int64_t hid, int64_t n_dc,
                                            int64_t n_g) {
  TORCH_CHECK(pw.is_cuda() && obs.is_cuda() && pw.dtype() == torch::kFloat32);
  int B = obs.size(0), D = obs.size(1);
  auto out_dc = torch::empty({B, n_dc}, obs.options());
  auto out_g = torch::empty({B, n_g}, obs.options());
  int rpb = REPLICAS_PER_BLOCK;
  int blocks = (B + rpb - 1) / rpb;
  size_t shmem = (size_t)rpb *
      (RL_MAX_OBS + 2 * RL_MAX_HID + RL_MAX_LOG) * sizeof(float);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rl_forward_kernel, dim3(blocks), dim3(THREADS_PER_BLOCK),
                     shmem, stream, pw.data_ptr<float>(),
                     obs.data_ptr<float>(), B, D, (int)hid, (int)n_dc,
                     (int)n_g, out_dc.data_ptr<float>(),
                     out_g.data_ptr<float>());
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("rl_forward_kernel: ") +
                             hipGetErrorString(e));
  return {out_dc, out_g};
}

// ---------------- host side ----------------
#define T_PTR(name, type) S_.name = t_[#name].data_ptr<type>()
#define T_CPTR(name, type) S_.name = t_[#name].data_ptr<type>()

class BatchedSimHip {
 public:
  BatchedSimHip(py::dict tensors, py::dict cfg) {
    // keep tensor refs alive
    for (auto item : tensors)
      t_[py::cast<std::string>(item.first)] = py::cast<torch::Tensor>(item.second);

    S_.n_rep = cfg["n_rep"].cast<int>();
    S_.n_dc = cfg["n_dc"].cast<int>();
    S_.n_ing = cfg["n_ing"].cast<int>();
    S_.n_freq = cfg["n_freq"].cast<int>();
    S_.n_streams = S_.n_ing * 2;
    S_.total_slots = cfg["total_slots"].cast<int>();
    S_.tcap = cfg["tcap"].cast<int>();
    S_.qcap = cfg["qcap"].cast<int>();
    S_.end_time = cfg["end_time"].cast<double>();
    S_.log_interval = cfg["log_interval"].cast<double>();
    S_.algo = cfg["algo"].cast<int>();
    S_.max_gpj = cfg["max_gpj"].cast<int>();
    S_.inf_priority = cfg["inf_priority"].cast<int>();
    S_.scale_out_low = cfg["scale_out_low"].cast<int>();
    S_.energy_aware = cfg["energy_aware"].cast<int>();
    S_.dvfs_low = cfg["dvfs_low"].cast<double>();
    S_.dvfs_high = cfg["dvfs_high"].cast<double>();
    S_.power_cap = cfg["power_cap"].cast<double>();
    S_.eco_obj = cfg["eco_obj"].cast<int>();
    S_.fp32_score = cfg.contains("fp32_score") ? cfg["fp32_score"].cast<int>() : 0;
    S_.num_fixed = cfg["num_fixed"].cast<int>();
    S_.fixed_freq = cfg["fixed_freq"].cast<double>();
    S_.payload_gb[0] = cfg["payload_inf_gb"].cast<double>();
    S_.payload_gb[1] = cfg["payload_trn_gb"].cast<double>();
    auto am = cfg["arr_mode"].cast<std::vector<int>>();
    auto ar = cfg["arr_rate"].cast<std::vector<double>>();
    auto aa = cfg["arr_amp"].cast<std::vector<double>>();
    auto ap = cfg["arr_period"].cast<std::vector<double>>();
    for (int j = 0; j < 2; ++j) {
      S_.arr_mode[j] = am[j]; S_.arr_rate[j] = ar[j];
      S_.arr_amp[j] = aa[j]; S_.arr_period[j] = ap[j];
    }
    S_.seed = cfg["seed"].cast<uint64_t>();
    S_.rep_id_offset = cfg["rep_id_offset"].cast<int64_t>();
    S_.log_replica = cfg["log_replica"].cast<int>();
    S_.cl_cap = cfg["cl_cap"].cast<int>();
    S_.jl_cap = cfg["jl_cap"].cast<int>();

    T_CPTR(freq_levels, double); T_CPTR(pc, double); T_CPTR(lc, double);
    T_CPTR(wan_lat, double); T_CPTR(wan_bw, double); T_CPTR(carbon, double);
    T_CPTR(price24, double); T_CPTR(total_gpus, int); T_CPTR(p_idle, double);
    T_CPTR(p_sleep, double); T_CPTR(p_peak, double); T_CPTR(pow_alpha, double);
    T_CPTR(power_gating, int); T_CPTR(default_freq, double);
    T_CPTR(slot_off, int); T_CPTR(slot_dc, int);
    T_PTR(now, double); T_PTR(next_log, double);
    S_.rng_ctr = reinterpret_cast<uint64_t*>(t_["rng_ctr"].data_ptr<int64_t>());
    T_PTR(jid_ctr, int); T_PTR(done, int); T_PTR(err, int);
    T_PTR(arr_next, double);
    T_PTR(busy, int); T_PTR(cur_freq, double); T_PTR(energy_j, double);
    T_PTR(util_time, double); T_PTR(util_begin, double); T_PTR(acc_unit, double);
    T_PTR(p_active, double); T_PTR(sum_tpt, double); T_PTR(n_running, int);
    T_PTR(dc_min_finish, double); T_PTR(dc_min_slot, int);
    T_PTR(s_finish, double); T_PTR(seq_ctr, int);
    S_.s_rec = reinterpret_cast<SRec*>(t_["s_rec"].data_ptr<double>());
    S_.s_gpus = reinterpret_cast<short*>(t_["s_gpus"].data_ptr<int16_t>());
    S_.s_jtype = reinterpret_cast<char*>(t_["s_jtype"].data_ptr<int8_t>());
    T_PTR(x_time, double);
    S_.x_rec = reinterpret_cast<XRec*>(t_["x_rec"].data_ptr<double>());
    T_PTR(q_head, int); T_PTR(q_len, int); T_PTR(q_pay, double);
    if (S_.algo == A_CAP_GREEDY) T_PTR(snap_f, double);
    T_PTR(q_netlat, float); T_PTR(q_jid, int);
    S_.q_ing = reinterpret_cast<char*>(t_["q_ing"].data_ptr<int8_t>());
    T_PTR(b_n, int); T_PTR(b_s, double);
    S_.b_t = reinterpret_cast<long long*>(t_["b_t"].data_ptr<int64_t>());
    S_.ev_count = reinterpret_cast<long long*>(t_["ev_count"].data_ptr<int64_t>());
    S_.jobs_done = reinterpret_cast<long long*>(t_["jobs_done"].data_ptr<int64_t>());
    S_.jobs_done_inf =
        reinterpret_cast<long long*>(t_["jobs_done_inf"].data_ptr<int64_t>());
    T_PTR(sum_lat, double); T_PTR(sum_lat_inf, double); T_PTR(sum_wait, double);
    T_PTR(cl_count, int); T_PTR(cl_rows, double);
    T_PTR(jl_count, int); T_PTR(jl_rows, double);

    // arrival-trace replay mode
    S_.trace_mode = cfg.contains("trace_mode") ? cfg["trace_mode"].cast<int>() : 0;
    S_.trace_cap = cfg.contains("trace_cap") ? cfg["trace_cap"].cast<int>() : 0;
    if (S_.trace_mode) {
      T_CPTR(trace_time, double);
      T_CPTR(trace_size, double);
      S_.trace_dc = reinterpret_cast<const char*>(t_["trace_dc"].data_ptr<int8_t>());
      T_PTR(trace_pos, int);
    }

    // CHSAC-AF extras
    S_.obs_dim = cfg.contains("obs_dim") ? cfg["obs_dim"].cast<int>() : 0;
    S_.sla_p99_ms = cfg.contains("sla_p99_ms") ? cfg["sla_p99_ms"].cast<double>() : 500.0;
    S_.tr_cap = cfg.contains("tr_cap") ? cfg["tr_cap"].cast<int>() : 0;
    S_.elastic = cfg.contains("elastic") ? cfg["elastic"].cast<int>() : 0;
    S_.pp_cap = cfg.contains("pp_cap") ? cfg["pp_cap"].cast<int>() : 0;
    S_.serve_device = cfg.contains("serve_device") ? cfg["serve_device"].cast<int>() : 0;
    S_.hid = cfg.contains("rl_hid") ? cfg["rl_hid"].cast<int>() : 256;
    S_.n_g = S_.max_gpj;
    S_.rl_det = cfg.contains("rl_det") ? cfg["rl_det"].cast<int>() : 0;
    S_.tr_limit = cfg.contains("tr_limit") ? cfg["tr_limit"].cast<int>() : 0;
    if (S_.serve_device) {
      if (S_.hid > RL_MAX_HID || S_.obs_dim > RL_MAX_OBS ||
          S_.n_dc > 8 || S_.n_g > 8)
        throw std::runtime_error("serve_device limits: hid<=256, obs<=64, "
                                 "n_dc<=8, n_g<=8");
      S_.pw = t_["policy_weights"].data_ptr<float>();
    }
    if (S_.algo == A_CHSAC) {
      T_PTR(req_flag, int); T_PTR(req_obs, float); T_PTR(req_mdc, int);
      T_PTR(req_mg, int); T_PTR(resp_dc, int); T_PTR(resp_g, int);
      T_PTR(pend_kind, int); T_PTR(pend_size, double); T_PTR(pend_netlat, float);
      T_PTR(pend_jid, int); T_PTR(pend_ing, int); T_PTR(pend_jt, int);
      T_PTR(pend_dc, int); T_PTR(pend_from_inf, int); T_PTR(pend_enq, double);
      T_PTR(slot_s0, float);
      T_PTR(pp_count, int); T_PTR(pp_cursor, int);
      T_PTR(pp_size, double); T_PTR(pp_done, double); T_PTR(pp_start, double);
      T_PTR(pp_netlat, float);
      T_PTR(pp_jid, int);
      S_.pp_ing = reinterpret_cast<unsigned char*>(t_["pp_ing"].data_ptr<uint8_t>());
      S_.pp_dc = reinterpret_cast<unsigned char*>(t_["pp_dc"].data_ptr<uint8_t>());
      S_.pp_pcount = reinterpret_cast<unsigned char*>(t_["pp_pcount"].data_ptr<uint8_t>());
      T_PTR(pp_s0, float);
      S_.pp_adc = reinterpret_cast<unsigned char*>(t_["pp_adc"].data_ptr<uint8_t>());
      S_.pp_ag = reinterpret_cast<unsigned char*>(t_["pp_ag"].data_ptr<uint8_t>());
      S_.pp_nrew = reinterpret_cast<unsigned char*>(t_["pp_nrew"].data_ptr<uint8_t>());
      S_.pp_has_rl = reinterpret_cast<unsigned char*>(t_["pp_has_rl"].data_ptr<uint8_t>());
      T_PTR(x_s0, float);
      T_PTR(lat_hist, int);
      S_.lat_count = reinterpret_cast<long long*>(t_["lat_count"].data_ptr<int64_t>());
      T_PTR(lat_sum, double);
      S_.exact_p99 = cfg.contains("exact_p99") ? cfg["exact_p99"].cast<int>() : 0;
      S_.p99_win = cfg.contains("p99_win") ? cfg["p99_win"].cast<int>() : 2048;
      if (S_.exact_p99) {
        T_PTR(p99_sorted, double);
        T_PTR(p99_ring, double);
      }
      T_PTR(tr_count, int);
      T_PTR(tr_s0, float); T_PTR(tr_s1, float);
      S_.tr_adc = reinterpret_cast<unsigned char*>(t_["tr_adc"].data_ptr<uint8_t>());
      S_.tr_ag = reinterpret_cast<unsigned char*>(t_["tr_ag"].data_ptr<uint8_t>());
      T_PTR(tr_r, float); T_PTR(tr_costs, float);
      S_.tr_mdc = reinterpret_cast<unsigned char*>(t_["tr_mdc"].data_ptr<uint8_t>());
      S_.tr_mg = reinterpret_cast<unsigned char*>(t_["tr_mg"].data_ptr<uint8_t>());
    }
  }

  // Create a CU-masked stream for the advance kernel, excluding the LAST
  // `n_reserved` CUs.  The advance kernel's blocks are persistent for a
  // whole cycle (each wave loops until its replica yields), so without a
  // mask they occupy every wave slot on the device and concurrent work
  // (the overlapped SAC train kernels) starves behind them — stream
  // priorities cannot help because no slot ever frees.  Masking carves a
  // small CU island the advance never touches; kernels on ordinary streams
  // land there immediately.
  void set_resident_cap(bool on) { (void)on; }  // negative result; no-op

  void enable_masked_stream(int n_reserved) {
    if (masked_stream_) return;
    reserved_cus_ = n_reserved;
    hipDeviceProp_t prop;
    int dev;
    (void)hipGetDevice(&dev);
    (void)hipGetDeviceProperties(&prop, dev);
    int n_cu = prop.multiProcessorCount;
    int words = (n_cu + 31) / 32;
    std::vector<uint32_t> mask(words, 0u);
    int usable = n_cu - n_reserved;
    if (usable < 1) usable = 1;
    for (int i = 0; i < usable; ++i) mask[i / 32] |= (1u << (i % 32));
    hipError_t e = hipExtStreamCreateWithCUMask(&masked_stream_,
                                                (uint32_t)words, mask.data());
    if (e != hipSuccess) {
      masked_stream_ = nullptr;  // fall back to the torch stream
      throw std::runtime_error(std::string("CU-mask stream: ") +
                               hipGetErrorString(e));
    }
  }

  bool masked_active() const { return masked_stream_ != nullptr; }

  bool advance_done() {
    hipStream_t s = masked_stream_ ? masked_stream_
                                   : (hipStream_t)at::hip::getCurrentHIPStream();
    return hipStreamQuery(s) == hipSuccess;
  }

  void advance_sync() {
    hipStream_t s = masked_stream_ ? masked_stream_
                                   : (hipStream_t)at::hip::getCurrentHIPStream();
    hipError_t e = hipStreamSynchronize(s);
    if (e != hipSuccess)
      throw std::runtime_error(std::string("advance_sync: ") +
                               hipGetErrorString(e));
  }

  // launch one advance chunk; returns immediately (stream-async)
  void advance(double t_target, int64_t max_ev) {
    int rpb = REPLICAS_PER_BLOCK;
    int blocks = (S_.n_rep + rpb - 1) / rpb;
    // dynamic LDS: per replica, Hot (+ s_finish/x_time mirrors at SUBW==64,
    // + actor-forward scratch for chsac)
    size_t hot_sz = (sizeof(Hot) + 15) & ~size_t(15);
    size_t per_rep = hot_sz;
    if (SUBW == 64)
      per_rep += (size_t)S_.total_slots * sizeof(double) +
                 (size_t)S_.tcap * sizeof(double);
    if (S_.algo == A_CHSAC)
      per_rep += (RL_MAX_OBS + 2 * RL_MAX_HID + RL_MAX_LOG) * sizeof(float);
    size_t shmem = rpb * per_rep;
    if (shmem > 64 * 1024)
      throw std::runtime_error(
          "scenario too large for the LDS-mirrored engine (total_slots + "
          "tcap exceed the 64 KiB dynamic-LDS budget per block)");
    dim3 grid(blocks), block(THREADS_PER_BLOCK);
    hipStream_t stream = masked_stream_
        ? masked_stream_ : (hipStream_t)at::hip::getCurrentHIPStream();
    switch (S_.algo) {
      case A_DEFAULT:
        hipLaunchKernelGGL(advance_kernel<A_DEFAULT>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_CAP_UNIFORM:
        hipLaunchKernelGGL(advance_kernel<A_CAP_UNIFORM>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_CAP_GREEDY:
        hipLaunchKernelGGL(advance_kernel<A_CAP_GREEDY>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_JOINT_NF:
        hipLaunchKernelGGL(advance_kernel<A_JOINT_NF>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_BANDIT:
        hipLaunchKernelGGL(advance_kernel<A_BANDIT>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_CARBON_COST:
        hipLaunchKernelGGL(advance_kernel<A_CARBON_COST>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_ECO_ROUTE:
        hipLaunchKernelGGL(advance_kernel<A_ECO_ROUTE>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_DEBUG:
        hipLaunchKernelGGL(advance_kernel<A_DEBUG>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      case A_CHSAC:
        hipLaunchKernelGGL(advance_kernel<A_CHSAC>, grid, block, shmem, stream, S_, t_target, max_ev); break;
      default:
        throw std::runtime_error("unsupported algo for HIP engine");
    }
    hipError_t e = hipGetLastError();
    if (e != hipSuccess)
      throw std::runtime_error(std::string("advance_kernel launch failed: ") +
                               hipGetErrorString(e));
  }

 private:
  EngineDesc S_;
  std::unordered_map<std::string, torch::Tensor> t_;
  hipStream_t masked_stream_ = nullptr;
  int reserved_cus_ = 0;
};

#undef T_PTR
#undef T_CPTR

}  // namespace dcg

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "Batched MI355X (gfx950) Monte-Carlo replica engine";
  // module_local: the 64-wide and 8-wide extensions register the
  // same C++ type in one process
  py::class_<dcg::BatchedSimHip>(m, "BatchedSimHip", py::module_local())
      .def(py::init<py::dict, py::dict>())
      .def("advance", &dcg::BatchedSimHip::advance,
           py::arg("t_target"), py::arg("max_events"))
      .def("enable_masked_stream", &dcg::BatchedSimHip::enable_masked_stream,
           py::arg("n_reserved_cus"))
      .def("masked_active", &dcg::BatchedSimHip::masked_active)
      .def("set_resident_cap", &dcg::BatchedSimHip::set_resident_cap,
           py::arg("on"))
      .def("advance_done", &dcg::BatchedSimHip::advance_done)
      .def("advance_sync", &dcg::BatchedSimHip::advance_sync);
  m.def("rl_forward_mfma", &dcg::rl_forward_mfma,
        "MFMA (matrix-core) batched actor forward "
        "(pw, obs[B,D], hid, n_dc, n_g) -> (logits_dc, logits_g)",
        py::arg("pw"), py::arg("obs"), py::arg("hid"), py::arg("n_dc"),
        py::arg("n_g"));
  m.def("rl_forward_debug", &dcg::rl_forward_debug,
        "batched actor forward via the in-kernel serving math "
        "(pw, obs[B,D], hid, n_dc, n_g) -> (logits_dc, logits_g)",
        py::arg("pw"), py::arg("obs"), py::arg("hid"), py::arg("n_dc"),
        py::arg("n_g"));
}
