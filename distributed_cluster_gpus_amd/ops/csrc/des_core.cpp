// Native scalar discrete-event simulator core (C++17, pybind11).
//
// Same event semantics as the Python oracle (engine/oracle.py), which itself
// reproduces the reference simulator (reference event loop:
// simcore/simulator_paper_multi.py:412-480).  Uses the CPython-compatible
// MT19937 (pyrng.hpp) and mirrors the oracle's RNG draw order, floating-point
// expression shapes (pow(f,3) like Python's f**3, accumulation order over
// insertion-ordered running-job lists) and CSV formats, so that seed-for-seed
// its cluster_log.csv / job_log.csv are byte-identical to the oracle's — and
// hence to the reference's.  Covers the 8 non-RL algorithms; chsac_af runs
// through the Python/torch path (engine/oracle.py, engine/batched.py).
//
// This is the framework's fast host engine (the reference is pure Python at
// ~4k events/s; this core runs the same workload 1-2 orders of magnitude
// faster on one core) and the staging ground for the batched MI355X engine's
// host-side logic.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstdio>
#include <deque>
#include <limits>
#include <queue>
#include <string>
#include <unordered_map>
#include <vector>

#include "pyrng.hpp"

namespace py = pybind11;

namespace dcg {

static constexpr double INF = std::numeric_limits<double>::infinity();

// ---------- analytic models (shapes must match Python bit-for-bit) ----------
static inline double gpu_power_w(double f, const double* c) {
  f = std::max(0.0, f);
  return c[0] * std::pow(f, 3.0) + c[1] * f + c[2];  // Python f**3 -> pow
}
static inline double task_power_w(int n, double f, const double* c) {
  n = std::max(0, n);
  return n * gpu_power_w(f, c);
}
static inline double unit_time_s(int n, double f, const double* c) {
  n = std::max(1, n);
  f = std::max(1e-9, f);
  if (n == 1) return c[0] + c[1] / f;
  return (c[0] + c[1] / f + c[2] * n) / n;
}

// ---------- arrivals ----------
enum class ArrMode { POISSON, SINUSOID, OFF };

struct Arrival {
  ArrMode mode;
  double rate, amp, period;

  double lambda_t(double t) const {
    if (mode == ArrMode::POISSON) return rate;
    if (mode == ArrMode::SINUSOID)
      return std::max(0.0, rate * (1.0 + amp * std::sin(2.0 * M_PI *
                      std::fmod(t, period) / period)));
    return 0.0;
  }
  double next_interarrival(double t, PyRandom& rng) const {
    if (mode == ArrMode::POISSON)
      return rate <= 0 ? INF : rng.expovariate(rate);
    if (mode == ArrMode::SINUSOID) {
      double max_rate = rate * (1.0 + std::fabs(amp));
      while (true) {
        double w = max_rate <= 0 ? INF : rng.expovariate(max_rate);
        // faithful non-accumulating thinning (reference arrivals.py:39-44)
        if (rng.random() <= lambda_t(t + w) / max_rate) return w;
      }
    }
    return INF;
  }
};

static inline double sample_job_size(int jtype, PyRandom& rng) {
  if (jtype == 0) {  // inference: Pareto(xm=1, alpha=1.8)
    double u = std::max(1e-9, 1.0 - rng.random());
    return 1.0 / std::pow(u, 1.0 / 1.8);
  }
  double v = rng.lognormvariate(std::log(50000.0), 0.4);
  return std::max(0.1, v);
}

// ---------- state ----------
struct Job {
  int64_t jid;
  int ing, jtype;  // jtype 0=inference 1=training
  double size, arrival_time;
  int dc = -1, gpus = 0;
  double start_time = 0, net_lat = 0, f_used = 0;
  double units_total = 0, units_done = 0, last_update = 0;
  int ev_gen = 0, preempt_count = 0;
};

struct DCState {
  double current_freq;
  int busy = 0;
  std::vector<int64_t> running;  // insertion-ordered (Python dict semantics)
  std::deque<int64_t> q_inf, q_train;
  double energy_j = 0, last_energy_t = 0;
  double util_gpu_time = 0, util_last = 0, util_begin = 0, acc_job_unit = 0;
};

enum EvType : int8_t { EV_ARR_INF = 0, EV_ARR_TRN = 1, EV_XFER = 2,
                       EV_FINISH = 3, EV_LOG = 4 };

struct Ev {
  double t;
  int64_t seq;
  int8_t type;
  int a;        // ingress (arrivals) / dc (finish)
  int64_t b;    // jid
  int c;        // gen
};
struct EvCmp {
  bool operator()(const Ev& x, const Ev& y) const {
    if (x.t != y.t) return x.t > y.t;
    return x.seq > y.seq;
  }
};

enum Algo { A_DEFAULT = 0, A_CAP_UNIFORM, A_CAP_GREEDY, A_JOINT_NF, A_BANDIT,
            A_CARBON_COST, A_ECO_ROUTE, A_DEBUG };

struct GridResult { int n; double f, T, P, E; };

class DesSim {
 public:
  DesSim(py::dict sc, py::dict params) : rng_(0) {
    // ---- scenario tables ----
    n_dc_ = sc["n_dc"].cast<int>();
    n_ing_ = sc["n_ing"].cast<int>();
    dc_names_ = sc["dc_names"].cast<std::vector<std::string>>();
    total_gpus_ = sc["total_gpus"].cast<std::vector<int>>();
    p_idle_ = sc["p_idle"].cast<std::vector<double>>();
    p_sleep_ = sc["p_sleep"].cast<std::vector<double>>();
    power_gating_ = sc["power_gating"].cast<std::vector<int>>();
    freq_levels_ = sc["freq_levels"].cast<std::vector<double>>();
    default_freq_ = sc["default_freq"].cast<std::vector<double>>();
    pc_ = sc["power_coeffs"].cast<std::vector<double>>();     // [dc*2*3]
    lc_ = sc["latency_coeffs"].cast<std::vector<double>>();
    wan_lat_ = sc["wan_latency_s"].cast<std::vector<double>>();   // [ing*dc]
    wan_bw_ = sc["wan_bottleneck_gbps"].cast<std::vector<double>>();
    carbon_ = sc["carbon"].cast<std::vector<double>>();
    price24_ = sc["price24"].cast<std::vector<double>>();
    payload_gb_[0] = sc["payload_inf_gb"].cast<double>();
    payload_gb_[1] = sc["payload_trn_gb"].cast<double>();
    // policy
    policy_energy_aware_ = sc["policy_name"].cast<std::string>() == "energy_aware";
    max_gpj_ = sc["max_gpus_per_job"].cast<int>();
    inf_priority_ = sc["inf_priority"].cast<bool>();
    dvfs_low_ = sc["dvfs_low"].cast<double>();
    dvfs_high_ = sc["dvfs_high"].cast<double>();
    scale_out_low_ = sc["train_scale_out_low_freq"].cast<bool>();

    // ---- run params ----
    algo_ = static_cast<Algo>(params["algo"].cast<int>());
    end_time_ = params["duration"].cast<double>();
    log_interval_ = params["log_interval"].cast<double>();
    rng_.seed(params["seed"].cast<uint64_t>());
    power_cap_ = params["power_cap"].cast<double>();
    eco_objective_ = params["eco_objective"].cast<int>();  // 0 energy 1 carbon 2 cost
    num_fixed_gpus_ = params["num_fixed_gpus"].cast<int>();
    fixed_freq_ = params["fixed_freq"].cast<double>();     // <=0 -> auto
    cluster_path_ = params["cluster_csv"].cast<std::string>();
    job_path_ = params["job_csv"].cast<std::string>();
    arr_[0] = parse_arrival(params["arr_inf"].cast<py::dict>());
    arr_[1] = parse_arrival(params["arr_trn"].cast<py::dict>());

    dcs_.resize(n_dc_);
    for (int d = 0; d < n_dc_; ++d) dcs_[d].current_freq = default_freq_[d];
    if (algo_ == A_BANDIT) {
      bandit_N_.assign(n_dc_ * 2 * freq_levels_.size(), 0);
      bandit_S_.assign(n_dc_ * 2 * freq_levels_.size(), 0.0);
    }
  }

  py::dict run() {
    auto t_start = std::chrono::steady_clock::now();
    fc_ = std::fopen(cluster_path_.c_str(), "wb");
    fj_ = std::fopen(job_path_.c_str(), "wb");
    if (!fc_ || !fj_) throw std::runtime_error("cannot open output CSVs");
    std::fprintf(fc_, "time_s,dc,freq,busy,free,run_total,run_inf,run_train,"
                      "q_inf,q_train,util_inst,util_avg,acc_job_unit,"
                      "power_W,energy_kJ\r\n");
    std::fprintf(fj_, "jid,ingress,type,size,dc,f_used,n_gpus,net_lat_s,"
                      "start_s,finish_s,latency_s,preempt_count,T_pred,"
                      "P_pred,E_pred\r\n");

    // seed arrivals (one inf + one trn per ingress, then first log tick)
    for (int i = 0; i < n_ing_; ++i) {
      schedule(now_ + arr_[0].next_interarrival(now_, rng_), EV_ARR_INF, i, 0, 0);
      schedule(now_ + arr_[1].next_interarrival(now_, rng_), EV_ARR_TRN, i, 0, 0);
    }
    schedule(now_ + log_interval_, EV_LOG, 0, 0, 0);

    while (!heap_.empty()) {
      Ev ev = heap_.top();
      heap_.pop();
      if (ev.t > end_time_) break;

      // per-event util + energy accrual for every DC, before dispatch
      for (int d = 0; d < n_dc_; ++d) {
        DCState& dc = dcs_[d];
        if (dc.util_last == 0.0) {
          dc.util_last = ev.t;
          dc.util_begin = ev.t;
        } else {
          double dt = std::max(0.0, ev.t - dc.util_last);
          dc.util_gpu_time += dc.busy * dt;
          dc.util_last = ev.t;
        }
        // accrue_energy (models.py:93-106 semantics)
        if (dc.last_energy_t == 0.0) {
          dc.last_energy_t = ev.t;
        } else {
          double dt = std::max(0.0, ev.t - dc.last_energy_t);
          dc.energy_j += dc_power(d) * dt;
          dc.last_energy_t = ev.t;
        }
      }

      now_ = ev.t;
      ++events_;
      switch (ev.type) {
        case EV_ARR_INF: on_arrival(0, ev.a); break;
        case EV_ARR_TRN: on_arrival(1, ev.a); break;
        case EV_XFER: on_transfer_done(ev.b); break;
        case EV_FINISH: {
          DCState& dc = dcs_[ev.a];
          auto it = jobs_.find(ev.b);
          if (it == jobs_.end()) break;
          Job& job = it->second;
          if (job.dc != ev.a) break;
          if (!is_running(dc, ev.b)) break;
          if (ev.c != job.ev_gen) break;  // stale (lazy invalidation)
          on_job_finish(ev.a, ev.b);
          break;
        }
        case EV_LOG: control(); on_log(); break;
      }
    }

    for (int d = 0; d < n_dc_; ++d) {
      DCState& dc = dcs_[d];
      if (0.0 < dc.util_last && dc.util_last < end_time_) {
        dc.util_gpu_time += dc.busy * (end_time_ - dc.util_last);
        dc.util_last = end_time_;
      }
      // final accrue_energy(end_time) without power_fn -> baseline model is
      // NOT used here: the reference calls accrue_energy(end) with no power_fn
      // which falls back to the baseline idle model -- but only when
      // last_energy_time != 0; replicate exactly:
      if (dc.last_energy_t == 0.0) {
        dc.last_energy_t = end_time_;
      } else {
        double dt = std::max(0.0, end_time_ - dc.last_energy_t);
        dc.energy_j += baseline_power(d) * dt;
        dc.last_energy_t = end_time_;
      }
    }

    std::fclose(fc_);
    std::fclose(fj_);
    auto t_end = std::chrono::steady_clock::now();
    double wall = std::chrono::duration<double>(t_end - t_start).count();
    double total_e = 0;
    for (auto& dc : dcs_) total_e += dc.energy_j;
    py::dict out;
    out["events"] = events_;
    out["wall_s"] = wall;
    out["events_per_sec"] = events_ / std::max(1e-12, wall);
    out["jobs_completed"] = jobs_completed_;
    out["total_energy_j"] = total_e;
    out["rl_updates"] = 0;
    return out;
  }

 private:
  // ---- scenario ----
  int n_dc_, n_ing_;
  std::vector<std::string> dc_names_;
  std::vector<int> total_gpus_;
  std::vector<double> p_idle_, p_sleep_;
  std::vector<int> power_gating_;
  std::vector<double> freq_levels_, default_freq_, pc_, lc_;
  std::vector<double> wan_lat_, wan_bw_, carbon_, price24_;
  double payload_gb_[2];
  bool policy_energy_aware_, inf_priority_, scale_out_low_;
  int max_gpj_;
  double dvfs_low_, dvfs_high_;

  // ---- run state ----
  Algo algo_;
  double end_time_, log_interval_, power_cap_, fixed_freq_;
  int eco_objective_, num_fixed_gpus_;
  std::string cluster_path_, job_path_;
  Arrival arr_[2];
  PyRandom rng_;
  double now_ = 0.0;
  int64_t seq_ = 0, jid_counter_ = 0, events_ = 0, jobs_completed_ = 0;
  std::priority_queue<Ev, std::vector<Ev>, EvCmp> heap_;
  std::vector<DCState> dcs_;
  std::unordered_map<int64_t, Job> jobs_;
  std::vector<int64_t> bandit_N_;
  std::vector<double> bandit_S_;
  int64_t bandit_t_ = 0;
  FILE* fc_ = nullptr;
  FILE* fj_ = nullptr;
  static constexpr double CAP_MARGIN = 5.0;

  static Arrival parse_arrival(py::dict d) {
    Arrival a;
    std::string m = d["mode"].cast<std::string>();
    a.mode = m == "poisson" ? ArrMode::POISSON
             : (m == "sinusoid" ? ArrMode::SINUSOID : ArrMode::OFF);
    a.rate = d["rate"].cast<double>();
    a.amp = d["amp"].cast<double>();
    a.period = d["period"].cast<double>();
    return a;
  }

  const double* pcoef(int d, int jtype) const { return &pc_[(d * 2 + jtype) * 3]; }
  const double* lcoef(int d, int jtype) const { return &lc_[(d * 2 + jtype) * 3]; }

  void schedule(double t, int8_t type, int a, int64_t b, int c) {
    if (t == INF || t > end_time_ + 1e-9) return;
    heap_.push(Ev{t, seq_++, type, a, b, c});
  }

  static bool is_running(const DCState& dc, int64_t jid) {
    return std::find(dc.running.begin(), dc.running.end(), jid) != dc.running.end();
  }
  void erase_running(DCState& dc, int64_t jid) {
    auto it = std::find(dc.running.begin(), dc.running.end(), jid);
    if (it != dc.running.end()) dc.running.erase(it);
  }

  // paper power model (insertion-order sum, matching Python accumulation)
  double dc_power(int d) const {
    const DCState& dc = dcs_[d];
    double p_active = 0.0;
    for (int64_t jid : dc.running) {
      const Job& job = jobs_.at(jid);
      p_active += task_power_w(job.gpus, job.f_used, pcoef(d, job.jtype));
    }
    int idle = total_gpus_[d] - dc.busy;
    double p_idle = idle * (power_gating_[d] ? p_sleep_[d] : p_idle_[d]);
    return p_active + p_idle;
  }

  // baseline (idle/sleep + f^alpha) model -- only used by the reference's
  // final accrue_energy(end_time) call with no power_fn (models.py:82-91).
  double baseline_power(int d) const {
    const DCState& dc = dcs_[d];
    double f = dc.current_freq;
    int active = dc.busy;
    int idle = total_gpus_[d] - active;
    // paper GPUSpec alpha is 3.0 for every type (configs/paper.py)
    double p_active = active * (p_idle_stub(d) + p_peak_stub(d) * std::pow(f, alpha_stub(d)));
    double p_idle = idle * (power_gating_[d] ? p_sleep_[d] : p_idle_stub(d));
    return p_active + p_idle;
  }
  double p_idle_stub(int d) const { return p_idle_[d]; }
  double p_peak_stub(int d) const { return p_peak_.empty() ? 0.0 : p_peak_[d]; }
  double alpha_stub(int d) const { return pow_alpha_.empty() ? 3.0 : pow_alpha_[d]; }

 public:
  // baseline-model extras (set from Python right after construction)
  std::vector<double> p_peak_, pow_alpha_;
  void set_baseline(std::vector<double> p_peak, std::vector<double> alpha) {
    p_peak_ = std::move(p_peak);
    pow_alpha_ = std::move(alpha);
  }

 private:
  double price_kwh() const {
    int h = static_cast<int>(std::fmod(now_, 86400.0) / 3600.0);
    return price24_[h];
  }

  GridResult best_nf_grid(int d, int jtype, int objective, double ci,
                          double price, bool has_ddl, double ddl) const {
    const double* pcf = pcoef(d, jtype);
    const double* tcf = lcoef(d, jtype);
    bool found = false;
    double best_score = 0;
    GridResult best{};
    for (int n = 1; n <= std::max(1, max_gpj_); ++n) {
      for (double f : freq_levels_) {
        double T = unit_time_s(n, f, tcf);
        double P = task_power_w(n, f, pcf);
        double E = P * T;
        if (has_ddl && T > ddl) continue;
        double score = E;
        if (objective == 1) score = E * ci;
        else if (objective == 2) score = (E / 3.6e6) * price;
        if (!found || score < best_score) {
          found = true;
          best_score = score;
          best = {n, f, T, P, E};
        }
      }
    }
    if (!found) {
      double fmax = *std::max_element(freq_levels_.begin(), freq_levels_.end());
      double T = unit_time_s(1, fmax, tcf);
      double P = gpu_power_w(fmax, pcf);  // per-GPU fallback (reference quirk)
      return {1, fmax, T, P, P * T};
    }
    return best;
  }

  double best_energy_freq(int n, int d, int jtype) const {
    const double* pcf = pcoef(d, jtype);
    const double* tcf = lcoef(d, jtype);
    bool found = false;
    double best_f = 0, best_e = 0;
    for (double f : freq_levels_) {
      double T = unit_time_s(n, f, tcf);
      double E = task_power_w(n, f, pcf) * T;
      if (!found || E < best_e) {
        found = true;
        best_e = E;
        best_f = f;
      }
    }
    return best_f;
  }

  // ---------- arrival / routing ----------
  void on_arrival(int jtype, int ing) {
    int64_t jid = ++jid_counter_;
    double size = sample_job_size(jtype, rng_);
    Job job;
    job.jid = jid;
    job.ing = ing;
    job.jtype = jtype;
    job.size = size;
    job.arrival_time = now_;

    int dc_idx;
    if (algo_ == A_ECO_ROUTE) {
      bool have = false;
      double best_score = 0;
      int best_d = 0;
      for (int d = 0; d < n_dc_; ++d) {
        double score = score_dc(d, jtype, size);
        if (!have || score < best_score) {
          have = true;
          best_score = score;
          best_d = d;
        }
      }
      dc_idx = best_d;
    } else {
      dc_idx = static_cast<int>(rng_.randbelow(static_cast<uint32_t>(n_dc_)));
    }
    double lnet = wan_lat_[ing * n_dc_ + dc_idx];
    double bw = wan_bw_[ing * n_dc_ + dc_idx];
    double xfer = (bw > 0.0) ? payload_gb_[jtype] / bw : 0.0;
    double transfer_s = lnet + xfer;
    job.dc = dc_idx;
    job.net_lat = lnet;
    jobs_.emplace(jid, job);
    schedule(now_ + transfer_s, EV_XFER, 0, jid, 0);

    double ia = arr_[jtype].next_interarrival(now_, rng_);
    schedule(now_ + ia, jtype == 0 ? EV_ARR_INF : EV_ARR_TRN, ing, 0, 0);
  }

  double score_dc(int d, int jtype, double size) const {
    // eco-route score (reference _score_dc_for_job, :1007-1039)
    if (eco_objective_ == 1) {
      double ci = carbon_[d];
      GridResult g = best_nf_grid(d, jtype, 1, ci, 0.0, false, 0.0);
      return (g.E * size) / 3.6e6 * ci;
    } else if (eco_objective_ == 2) {
      double price = price_kwh();
      GridResult g = best_nf_grid(d, jtype, 2, 0.0, price, false, 0.0);
      return (g.E * size) / 3.6e6 * price;
    }
    GridResult g = best_nf_grid(d, jtype, 0, 0.0, 0.0, false, 0.0);
    return g.E * size;
  }

  // ---------- DC admission ----------
  void on_transfer_done(int64_t jid) {
    Job& job = jobs_.at(jid);
    int d = job.dc;
    DCState& dc = dcs_[d];
    job.arrival_time = now_;
    if (total_gpus_[d] - dc.busy > 0) {
      if (decide_and_start(d, jid)) return;
    }
    (job.jtype == 0 ? dc.q_inf : dc.q_train).push_back(jid);
  }

  bool decide_and_start(int d, int64_t jid) {
    DCState& dc = dcs_[d];
    Job& job = jobs_.at(jid);
    int free = total_gpus_[d] - dc.busy;
    switch (algo_) {
      case A_JOINT_NF: {
        GridResult g = best_nf_grid(d, job.jtype, 0, 0.0, 0.0, false, 0.0);
        start_with_nf(d, jid, g.n, g.f);
        return true;
      }
      case A_BANDIT: {
        int n = std::min(free, max_gpj_);
        double f = bandit_select(d, job.jtype);
        start_with_nf(d, jid, n, f);
        return true;
      }
      case A_CARBON_COST: {
        double price = price_kwh();
        GridResult g = (price > 0.0)
            ? best_nf_grid(d, job.jtype, 2, 0.0, price, false, 0.0)
            : best_nf_grid(d, job.jtype, 1, carbon_[d], 0.0, false, 0.0);
        start_with_nf(d, jid, g.n, g.f);
        return true;
      }
      case A_DEBUG: {
        int n = num_fixed_gpus_;
        double f = fixed_freq_ > 0 ? fixed_freq_ : best_energy_freq(n, d, job.jtype);
        start_with_nf(d, jid, n, f);
        return true;
      }
      default: {  // heuristic (default_policy, cap_*, eco_route)
        int g = heuristic_allocate(d, job.jtype);
        if (g > 0) {
          start_heuristic(d, jid, g);
          return true;
        }
        return false;
      }
    }
  }

  int heuristic_allocate(int d, int jtype) {
    // reference select_gpus_and_set_freq (policy.py:16-41)
    DCState& dc = dcs_[d];
    int free = total_gpus_[d] - dc.busy;
    int g = free > 0 ? std::min(free, max_gpj_) : 0;
    if (!policy_energy_aware_) {  // perf_first
      if (jtype == 0) {
        dc.current_freq = dvfs_high_;
        return std::max(1, g);
      }
      dc.current_freq = std::max(dc.current_freq,
                                 !dc.q_inf.empty() ? dvfs_high_ : default_freq_[d]);
      return std::max(1, g);
    }
    if (jtype == 0) {
      dc.current_freq = dvfs_high_;
      return std::max(1, g);
    }
    if (scale_out_low_ && free >= 2) {
      dc.current_freq = dvfs_low_;
      g = std::min(free, max_gpj_);
      return std::max(1, g);
    }
    dc.current_freq = std::max(dc.current_freq, dvfs_low_);
    return std::max(1, g);
  }

  void start_heuristic(int d, int64_t jid, int gpus) {
    DCState& dc = dcs_[d];
    Job& job = jobs_.at(jid);
    if (gpus <= 0) {
      (job.jtype == 0 ? dc.q_inf : dc.q_train).push_back(jid);
      return;
    }
    dc.busy += gpus;
    dc.running.push_back(jid);
    job.gpus = gpus;
    job.start_time = now_;
    job.f_used = dc.current_freq;
    job.units_total = job.size;
    job.units_done = 0.0;
    job.last_update = now_;
    job.ev_gen += 1;
    double T = unit_time_s(gpus, dc.current_freq, lcoef(d, job.jtype));
    schedule(now_ + job.size * T, EV_FINISH, d, jid, job.ev_gen);
  }

  void start_with_nf(int d, int64_t jid, int n, double f) {
    DCState& dc = dcs_[d];
    Job& job = jobs_.at(jid);
    n = std::max(1, std::min(n, total_gpus_[d] - dc.busy));
    if (n <= 0) {
      (job.jtype == 0 ? dc.q_inf : dc.q_train).push_back(jid);
      return;
    }
    dc.busy += n;
    dc.running.push_back(jid);
    job.gpus = n;
    job.start_time = now_;
    job.f_used = f;
    job.units_total = job.size;
    job.units_done = 0.0;
    job.last_update = now_;
    job.ev_gen += 1;
    double T = unit_time_s(n, f, lcoef(d, job.jtype));
    schedule(now_ + job.size * T, EV_FINISH, d, jid, job.ev_gen);
  }

  double bandit_select(int d, int jtype) {
    // UCB1 (reference learners.py:20-36)
    bandit_t_ += 1;
    size_t nf = freq_levels_.size();
    for (size_t k = 0; k < nf; ++k) {
      if (bandit_N_[(d * 2 + jtype) * nf + k] < 1) return freq_levels_[k];
    }
    double best_f = 0, best_ucb = -1e9;
    bool have = false;
    for (size_t k = 0; k < nf; ++k) {
      int64_t n = bandit_N_[(d * 2 + jtype) * nf + k];
      double mean = n > 0 ? bandit_S_[(d * 2 + jtype) * nf + k] / n : 0.0;
      double ucb = mean + std::sqrt(2.0 * std::log((double)bandit_t_) / n);
      if (!have || ucb > best_ucb) {
        have = true;
        best_ucb = ucb;
        best_f = freq_levels_[k];
      }
    }
    return best_f;
  }

  void bandit_update(int d, int jtype, double f, double cost) {
    size_t nf = freq_levels_.size();
    for (size_t k = 0; k < nf; ++k) {
      if (freq_levels_[k] == f) {
        bandit_N_[(d * 2 + jtype) * nf + k] += 1;
        bandit_S_[(d * 2 + jtype) * nf + k] += -cost;
        return;
      }
    }
  }

  // ---------- completion ----------
  void on_job_finish(int d, int64_t jid) {
    DCState& dc = dcs_[d];
    Job job = jobs_.at(jid);  // copy; we erase below
    erase_running(dc, jid);
    dc.busy = std::max(0, dc.busy - job.gpus);
    double finish_time = now_;
    ++jobs_completed_;

    // remainder job-units: window = finish_time mod log_interval (quirk kept)
    accumulate_job_unit(d, job, std::fmod(finish_time, log_interval_));

    const double* pcf = pcoef(d, job.jtype);
    const double* tcf = lcoef(d, job.jtype);
    double T_pred = unit_time_s(job.gpus, job.f_used, tcf);
    double P_pred = task_power_w(job.gpus, job.f_used, pcf);
    double E_pred = P_pred * T_pred;

    std::fprintf(fj_, "%lld,%s,%s,%.4f,%s,%.3f,%d,%.4f,%.6f,%.6f,%.6f,%d,"
                      "%.6f,%.2f,%.2f\r\n",
                 (long long)job.jid, ing_name(job.ing).c_str(),
                 job.jtype == 0 ? "inference" : "training", job.size,
                 dc_names_[d].c_str(), job.f_used, job.gpus, job.net_lat,
                 job.start_time, finish_time, finish_time - job.start_time,
                 job.preempt_count, T_pred, P_pred, E_pred);

    if (algo_ == A_BANDIT) bandit_update(d, job.jtype, job.f_used, E_pred);

    jobs_.erase(jid);
    drain_queues(d);
  }

  void accumulate_job_unit(int d, const Job& job, double window) {
    double tpt = 1.0 / unit_time_s(job.gpus, job.f_used, lcoef(d, job.jtype));
    dcs_[d].acc_job_unit += tpt * window;
  }

  void drain_queues(int d) {
    DCState& dc = dcs_[d];
    while (total_gpus_[d] - dc.busy > 0) {
      int64_t jid = -1;
      bool from_inf = false;
      if (inf_priority_ && !dc.q_inf.empty()) {
        jid = dc.q_inf.front();
        dc.q_inf.pop_front();
        from_inf = true;
      } else if (!dc.q_train.empty()) {
        jid = dc.q_train.front();
        dc.q_train.pop_front();
      }
      if (jid < 0) break;
      Job& job = jobs_.at(jid);
      switch (algo_) {
        case A_JOINT_NF: {
          GridResult g = best_nf_grid(d, job.jtype, 0, 0.0, 0.0, false, 0.0);
          start_with_nf(d, jid, g.n, g.f);
          break;
        }
        case A_BANDIT: {
          int n = std::min(total_gpus_[d] - dc.busy, max_gpj_);
          double f = bandit_select(d, job.jtype);
          start_with_nf(d, jid, n, f);
          break;
        }
        case A_CARBON_COST: {
          GridResult g = best_nf_grid(d, job.jtype, 1, carbon_[d], 0.0, false, 0.0);
          start_with_nf(d, jid, g.n, g.f);
          break;
        }
        default: {
          int g = heuristic_allocate(d, job.jtype);
          if (g <= 0) {
            (from_inf ? dc.q_inf : dc.q_train).push_front(jid);
            return;
          }
          start_heuristic(d, jid, g);
          break;
        }
      }
    }
  }

  // ---------- log tick + power-cap control ----------
  void on_log() {
    for (int d = 0; d < n_dc_; ++d) {
      DCState& dc = dcs_[d];
      int run_total = static_cast<int>(dc.running.size());
      int run_inf = 0;
      for (int64_t jid : dc.running)
        if (jobs_.at(jid).jtype == 0) ++run_inf;
      int run_trn = run_total - run_inf;
      double util_inst = total_gpus_[d] ? (double)dc.busy / total_gpus_[d] : 0.0;
      double begin = dc.util_begin != 0.0 ? dc.util_begin : now_;
      double elapsed = std::max(1e-9, now_ - begin);
      double util_avg = total_gpus_[d]
          ? dc.util_gpu_time / (total_gpus_[d] * elapsed) : 0.0;
      double power_now = dc_power(d);
      for (int64_t jid : dc.running)
        accumulate_job_unit(d, jobs_.at(jid), log_interval_);
      std::fprintf(fc_, "%.3f,%s,%.2f,%d,%d,%d,%d,%d,%d,%d,%.4f,%.4f,%.4f,"
                        "%.2f,%.4f\r\n",
                   now_, dc_names_[d].c_str(), dc.current_freq, dc.busy,
                   total_gpus_[d] - dc.busy, run_total, run_inf, run_trn,
                   (int)dc.q_inf.size(), (int)dc.q_train.size(),
                   util_inst, util_avg, dc.acc_job_unit, power_now,
                   dc.energy_j / 1000.0);
    }
    schedule(now_ + log_interval_, EV_LOG, 0, 0, 0);
  }

  void control() {
    if (power_cap_ <= 0) return;
    if (algo_ != A_CAP_UNIFORM && algo_ != A_CAP_GREEDY) {
      if (algo_ == A_ECO_ROUTE || algo_ == A_CARBON_COST) {
        for (int d = 0; d < n_dc_; ++d)
          if (dcs_[d].busy == 0 && !freq_levels_.empty())
            dcs_[d].current_freq =
                *std::min_element(freq_levels_.begin(), freq_levels_.end());
      }
      return;
    }
    double totalP = 0;
    for (int d = 0; d < n_dc_; ++d) totalP += dc_power(d);
    if (totalP <= power_cap_ - CAP_MARGIN) return;
    double deficit = std::max(0.0, totalP - power_cap_);
    if (deficit <= 1e-6) return;
    if (algo_ == A_CAP_UNIFORM) {
      // cap_uniform probes delta-P against a DC-level frequency step, but the
      // power model reads per-job f_used, so every delta is 0 and the loop
      // exits immediately (reference :181-205 behaviour preserved).
      return;
    }
    cap_greedy(deficit);
  }

  struct Atom { double rho, dV, dP; int64_t jid; int dc; double f_from, f_to; };

  void cap_greedy(double deficit) {
    int guard = 10000;
    std::vector<double> lv(freq_levels_);
    std::sort(lv.begin(), lv.end());
    double f_min = lv.front();
    while (deficit > 1e-6 && guard-- > 0) {
      // collect tasks + build down-atoms (freq_load_agg.py:44-80)
      std::vector<Atom> down;
      bool any_task = false;
      for (int d = 0; d < n_dc_; ++d) {
        DCState& dc = dcs_[d];
        for (int64_t jid : dc.running) {
          Job& job = jobs_.at(jid);
          double cur_f = job.f_used != 0.0 ? job.f_used : dc.current_freq;
          if (cur_f <= f_min + 1e-12) continue;
          any_task = true;
          const double* pcf = pcoef(d, job.jtype);
          const double* tcf = lcoef(d, job.jtype);
          // nearest ladder index to cur_f
          size_t i0 = 0;
          double bd = 1e300;
          for (size_t k = 0; k < lv.size(); ++k) {
            double diff = std::fabs(lv[k] - cur_f);
            if (diff < bd) { bd = diff; i0 = k; }
          }
          double curV = throughput(job.gpus, lv[i0], tcf);
          double curP = task_power_w(job.gpus, lv[i0], pcf);
          for (size_t k = i0; k > 0; --k) {
            double f_from = lv[k], f_to = lv[k - 1];
            double V2 = throughput(job.gpus, f_to, tcf);
            double P2 = task_power_w(job.gpus, f_to, pcf);
            double dV = std::max(0.0, curV - V2);
            double dP = std::max(0.0, curP - P2);
            if (dV > 0 && dP >= 0)
              down.push_back(Atom{dP / dV, dV, dP, jid, d, f_from, f_to});
            curV = V2;
            curP = P2;
          }
        }
      }
      if (!any_task || down.empty()) break;
      std::stable_sort(down.begin(), down.end(),
                       [](const Atom& a, const Atom& b) { return a.rho < b.rho; });
      bool applied = false;
      for (const Atom& atom : down) {
        if (deficit <= 1e-6) break;
        DCState& dc = dcs_[atom.dc];
        if (!is_running(dc, atom.jid)) continue;
        Job& job = jobs_.at(atom.jid);
        double cur_f = job.f_used != 0.0 ? job.f_used : dc.current_freq;
        if (atom.f_to >= cur_f - 1e-12) continue;
        reschedule_job(atom.dc, atom.jid, atom.f_to);
        applied = true;
        double totalP = 0;
        for (int d2 = 0; d2 < n_dc_; ++d2) totalP += dc_power(d2);
        deficit = std::max(0.0, totalP - power_cap_);
        if (deficit <= 1e-6) break;
      }
      if (!applied) break;
    }
  }

  static double throughput(int n, double f, const double* tcf) {
    double T = unit_time_s(n, f, tcf);
    return T <= 0 ? 0.0 : 1.0 / T;
  }

  void reschedule_job(int d, int64_t jid, double new_f) {
    DCState& dc = dcs_[d];
    Job& job = jobs_.at(jid);
    // advance progress at current f (simulator_paper_multi.py:323-338)
    double f_cur = job.f_used != 0.0 ? job.f_used : dc.current_freq;
    double rate = 1.0 / std::max(unit_time_s(job.gpus, f_cur, lcoef(d, job.jtype)), 1e-9);
    double dt = std::max(0.0, now_ - job.last_update);
    job.units_done = std::min(job.units_total, job.units_done + rate * dt);
    job.last_update = now_;
    job.f_used = new_f;
    double units_left = std::max(0.0, job.units_total - job.units_done);
    double rate_new = 1.0 / std::max(unit_time_s(job.gpus, new_f, lcoef(d, job.jtype)), 1e-9);
    double finish_in = units_left / std::max(rate_new, 1e-9);
    job.ev_gen += 1;
    schedule(now_ + finish_in, EV_FINISH, d, jid, job.ev_gen);
  }

  std::string ing_name(int i) const { return ing_names_.empty() ? "" : ing_names_[i]; }

 public:
  std::vector<std::string> ing_names_;
};

}  // namespace dcg

PYBIND11_MODULE(_des_core, m) {
  m.doc() = "Native scalar DES core (CPython-RNG-compatible)";
  py::class_<dcg::DesSim>(m, "DesSim")
      .def(py::init<py::dict, py::dict>())
      .def("set_baseline", &dcg::DesSim::set_baseline)
      .def("set_ingress_names",
           [](dcg::DesSim& s, std::vector<std::string> names) {
             s.ing_names_ = std::move(names);
           })
      .def("run", &dcg::DesSim::run);

  py::class_<dcg::PyRandom>(m, "PyRandom")
      .def(py::init<uint64_t>())
      .def("random", &dcg::PyRandom::random)
      .def("getrandbits", &dcg::PyRandom::getrandbits)
      .def("randbelow", &dcg::PyRandom::randbelow)
      .def("expovariate", &dcg::PyRandom::expovariate)
      .def("normalvariate", &dcg::PyRandom::normalvariate)
      .def("lognormvariate", &dcg::PyRandom::lognormvariate);
}
