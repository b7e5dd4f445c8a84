"""Scalar discrete-event oracle engine (pure Python).

This engine reproduces the reference simulator's event semantics exactly —
event ordering (heap keyed on (t, seq)), per-event energy/util accrual, lazy
invalidation of stale job_finish events via ev_gen, per-algorithm dispatch,
RNG draw order — so that, seed-for-seed, its cluster_log.csv / job_log.csv are
byte-identical to the reference's (reference event loop:
simcore/simulator_paper_multi.py:412-480; handlers :537-980).

It exists as (a) the correctness oracle every faster engine is verified
against (native C++ DES: bitwise; batched MI355X engine: replica-0 trajectory
and distributional checks), and (b) the CPU fallback path.

Engine-behaviour notes (quirks preserved deliberately):
* cap_uniform is a no-op under per-job DVFS: its delta-P probe evaluates DC
  power from each job's f_used, which a DC-level frequency step does not
  change, so every candidate delta is 0 and the loop exits immediately
  (reference :181-205).  Kept bit-for-bit; cap_greedy is the functional cap.
* The job-unit remainder credited at completion uses finish_time MOD
  log_interval (reference :711).
* chsac_af's queue drain starts at most one queued job per finish (:890).
"""
import heapq
import itertools
import os
import random
from typing import Dict, List, Optional, Tuple

from ..models.arrivals import ArrivalProcess, sample_job_size
from ..models.cluster import DataCenterState, JobState, PreemptedJobState
from ..models.latency import unit_time_s
from ..models.scenario import PAYLOAD_GB, Scenario
from ..policies.bandit import UCB1DVFS
from ..policies.gridsearch import best_energy_freq, best_nf_grid, energy_tuple
from ..policies.heuristic import heuristic_allocate
from ..policies.powercap import RunningTask, aggregate_atoms
from ..utils.csvlog import ClusterLogWriter, JobLogWriter
from ..utils.timers import ThroughputMeter

ALGOS = ("default_policy", "cap_uniform", "cap_greedy", "joint_nf", "bandit",
         "carbon_cost", "eco_route", "chsac_af", "debug")

J_PER_KWH = 3.6e6


class _NullLogger:
    def info(self, *a, **k): pass
    def debug(self, *a, **k): pass
    def warning(self, *a, **k): pass


class OracleEngine:
    def __init__(self, scenario: Scenario,
                 arrival_inf: ArrivalProcess, arrival_trn: ArrivalProcess,
                 *, algo: str = "default_policy",
                 duration: float = 3600.0, log_interval: float = 10.0,
                 out_dir: Optional[str] = None, seed: int = 42,
                 power_cap: float = 0.0, control_interval: float = 5.0,
                 use_control_interval: bool = False,
                 elastic_scaling: bool = False, eco_objective: str = "energy",
                 num_fixed_gpus: int = 1, fixed_freq: Optional[float] = None,
                 sla_p99_ms: float = 500.0, energy_budget_j: Optional[float] = None,
                 rl_device: str = "cpu", rl_batch: int = 256, rl_warmup: int = 1000,
                 rl_buffer: int = 200000, rl_agent=None,
                 logger=None, show_progress: bool = False,
                 cluster_writer=None, job_writer=None,
                 arrival_recorder=None, rl_deterministic: bool = False):
        if algo not in ALGOS:
            raise ValueError(f"unknown algo {algo!r}")
        self.sc = scenario
        self.algo = algo
        self.now = 0.0
        self.end_time = float(duration)
        self.log_interval = float(log_interval)
        self.power_cap = float(power_cap)
        self.control_interval = float(control_interval)
        # Reference parity: the cap controller fires on every LOG tick and
        # --control-interval is parsed but unused.
        # use_control_interval=True opts into a dedicated control cadence.
        self.use_control_interval = bool(use_control_interval)
        self.eco_objective = eco_objective
        self.num_fixed_gpus = int(num_fixed_gpus)
        self.fixed_freq = fixed_freq
        self.sla_p99_ms = float(sla_p99_ms)
        self.energy_budget_j = energy_budget_j
        self.logger = logger or _NullLogger()
        self.show_progress = show_progress
        # optional list collecting (t, ingress, jtype, size) per arrival —
        # used to build exact-replay traces for the GPU engine
        self.arrival_recorder = arrival_recorder
        # parity mode: greedy (argmax) policy actions so a frozen agent's
        # trajectory is reproducible across engines
        self.rl_deterministic = bool(rl_deterministic)
        self.elastic_scaling = bool(elastic_scaling) and algo == "chsac_af"

        # CPython Mersenne stream: seeded exactly like the reference
        # (random.seed(seed), simulator_paper_multi.py:71) but on a private
        # Random instance so concurrent engines don't share state.
        self.rng = random.Random(seed)
        self.arr_inf = arrival_inf
        self.arr_trn = arrival_trn

        self.dcs: Dict[str, DataCenterState] = scenario.make_dc_states()
        self._dc_names = list(self.dcs.keys())
        self._dc_idx = {n: i for i, n in enumerate(self._dc_names)}
        self._ing_idx = {n: i for i, n in enumerate(scenario.ingress_names)}
        # coefficient cache: (dc, jtype) -> (PowerCoeffs, LatencyCoeffs, raw
        # power triple); identical values to the scenario tables, avoids
        # rebuilding dataclasses in the per-event hot loop
        from ..models.coeffs import LatencyCoeffs, PowerCoeffs
        self._coeff_cache = {}
        for d, name in enumerate(self._dc_names):
            for j, jname in enumerate(("inference", "training")):
                pc = PowerCoeffs(*scenario.power_coeffs[d, j, :])
                tc = LatencyCoeffs(*scenario.latency_coeffs[d, j, :])
                self._coeff_cache[(name, jname)] = (pc, tc)
        self.event_q: List[Tuple[float, int, str, dict]] = []
        self._seq = itertools.count()
        self._jid = itertools.count(1)

        self.bandit = UCB1DVFS(init_explore=1, objective="energy") if algo == "bandit" else None
        self.cap_margin = 5.0  # W of hysteresis around the cap (reference :235)

        # tail-latency sliding windows per job type (reference :728-737)
        self._lat_hist: Dict[str, List[float]] = {"inference": [], "training": []}

        # === RL (chsac_af) ===
        self.rl = None
        self.replay = None
        self._rl_batch, self._rl_warmup = int(rl_batch), int(rl_warmup)
        if algo == "chsac_af":
            from ..rl.agent import CHSACAgentConfig, make_agent
            from ..rl.replay import ReplayRing
            f_all = [float(f) for f in scenario.freq_levels]
            self._f_min = min(f_all) if f_all else 1.0
            obs_dim = 1 + 6 * len(self._dc_names)
            constraints = {"latency_p99": self.sla_p99_ms}
            if self.power_cap > 0:
                constraints["power"] = float(self.power_cap)
            if energy_budget_j and energy_budget_j > 0:
                constraints["energy_total"] = float(energy_budget_j)
            constraints["gpu_over"] = 0.0
            if rl_agent is not None:
                self.rl = rl_agent
            else:
                self.rl = make_agent(CHSACAgentConfig(
                    obs_dim=obs_dim, n_dc=len(self._dc_names),
                    n_g_choices=int(scenario.policy.max_gpus_per_job),
                    constraints=constraints, device=rl_device))
            self.replay = ReplayRing(capacity=int(rl_buffer), obs_dim=obs_dim,
                                     n_costs=len(constraints),
                                     cost_names=list(constraints.keys()),
                                     n_dc=len(self._dc_names),
                                     n_g=int(scenario.policy.max_gpus_per_job),
                                     device=rl_device)

        # output writers
        self.cluster_writer = cluster_writer
        self.job_writer = job_writer
        if out_dir is not None:
            os.makedirs(out_dir, exist_ok=True)
            self.cluster_writer = ClusterLogWriter(os.path.join(out_dir, "cluster_log.csv"))
            self.job_writer = JobLogWriter(os.path.join(out_dir, "job_log.csv"))

        self.meter = ThroughputMeter()
        self.rl_updates = 0
        self.jobs_completed = 0

        # seed arrivals: one inference + one training event per ingress, in
        # ingress order, then the first log tick (reference :153-157).
        for ing_name in scenario.ingress_names:
            self._schedule(self.now + self.arr_inf.next_interarrival(self.now, self.rng),
                           "arrival_inf", {"ing": ing_name})
            self._schedule(self.now + self.arr_trn.next_interarrival(self.now, self.rng),
                           "arrival_trn", {"ing": ing_name})
        self._schedule(self.now + self.log_interval, "log", {"interval": self.log_interval})
        if self.use_control_interval and self.power_cap > 0:
            self._schedule(self.now + self.control_interval, "control", {})

    # ---------- event plumbing ----------
    def _schedule(self, t: float, etype: str, payload: dict):
        if t == float("inf") or t > self.end_time + 1e-9:
            return
        heapq.heappush(self.event_q, (t, next(self._seq), etype, payload))

    # ---------- power model ----------
    def _dc_power_w(self, dc: DataCenterState) -> float:
        """Paper power model: sum of per-job n*P(f_used) + sleeping/idle floor
        (reference _estimate_dc_power, :168-179)."""
        cache = self._coeff_cache
        name = dc.name
        p_active = 0.0
        for job, g in dc.running_jobs.values():
            pc = cache[(name, job.jtype)][0]
            f = max(0.0, job.f_used)
            p_active += max(0, int(g)) * (pc.alpha_p * f ** 3 + pc.beta_p * f
                                          + pc.gamma_p)
        idle = dc.total_gpus - dc.busy_gpus
        p_idle = idle * (dc.p_sleep if dc.power_gating else dc.p_idle)
        return p_active + p_idle

    def _coeffs(self, dc_name: str, jtype: str):
        return self._coeff_cache[(dc_name, jtype)]

    # ---------- run loop ----------
    def run(self):
        self.meter.start()
        pbar = None
        if self.show_progress:
            try:
                from tqdm.auto import tqdm
                pbar = tqdm(total=self.end_time, desc="Sim time", unit="s",
                            dynamic_ncols=True, mininterval=0.2)
            except Exception:
                pbar = None
        pbar_last = 0.0

        while self.event_q:
            t, _, etype, payload = heapq.heappop(self.event_q)
            if t > self.end_time:
                break

            # per-event util + energy accrual across all DCs, BEFORE dispatch
            # (reference :429-437)
            for dc in self.dcs.values():
                if dc.util_last_ts == 0.0:
                    dc.util_last_ts = t
                    dc.util_begin_ts = t
                else:
                    dt = max(0.0, t - dc.util_last_ts)
                    dc.util_gpu_time += dc.busy_gpus * dt
                    dc.util_last_ts = t
                dc.accrue_energy(t, power_fn=self._dc_power_w)

            if pbar is not None and t > pbar_last:
                pbar.update(t - pbar_last)
                pbar_last = t

            self.now = t
            self.meter.add(1)
            if etype == "arrival_inf":
                self._on_arrival("inference", payload["ing"])
            elif etype == "arrival_trn":
                self._on_arrival("training", payload["ing"])
            elif etype == "xfer_done":
                self._on_transfer_done(payload)
            elif etype == "job_finish":
                dc = self.dcs[payload["dc"]]
                tup = dc.running_jobs.get(payload["jid"])
                if not tup:
                    continue
                job, _ = tup
                if payload.get("gen") != job.ev_gen:
                    continue  # stale finish event (lazy invalidation)
                self._on_job_finish(payload["dc"], payload["jid"])
            elif etype == "control":
                self._control()
                self._schedule(self.now + self.control_interval, "control", {})
            elif etype == "log":
                if not self.use_control_interval:
                    self._control()
                self._on_log(payload["interval"])
            else:
                raise RuntimeError(f"Unknown event {etype}")

        # final flush to end_time (reference :469-475)
        for dc in self.dcs.values():
            if 0.0 < dc.util_last_ts < self.end_time:
                dc.util_gpu_time += dc.busy_gpus * (self.end_time - dc.util_last_ts)
                dc.util_last_ts = self.end_time
            dc.accrue_energy(self.end_time)

        if pbar is not None:
            if pbar.n < pbar.total:
                pbar.update(pbar.total - pbar.n)
            pbar.close()
        self.meter.stop()
        for w in (self.cluster_writer, self.job_writer):
            if w is not None:
                w.close()
        return self.stats()

    def stats(self):
        return {
            "events": self.meter.count,
            "wall_s": self.meter.elapsed_s,
            "events_per_sec": self.meter.per_sec,
            "rl_updates": self.rl_updates,
            "jobs_completed": self.jobs_completed,
            "total_energy_j": sum(dc.energy_joules for dc in self.dcs.values()),
        }

    # ---------- WAN ----------
    def _net_tuple(self, ing_name: str, dc_name: str, job: JobState):
        """(Lnet_s, bottleneck_gbps, cost_per_gb, transfer_s) from precomputed
        all-pairs tables (graph is static; reference runs Dijkstra per arrival,
        :482-496 — identical numbers)."""
        i = self._ing_idx[ing_name]
        d = self._dc_idx[dc_name]
        lnet = float(self.sc.wan_latency_s[i][d])
        bw = float(self.sc.wan_bottleneck_gbps[i][d])
        cost = float(self.sc.wan_cost_per_gb[i][d])
        data_gb = PAYLOAD_GB[0] if job.jtype == "inference" else PAYLOAD_GB[1]
        xfer = data_gb / bw if bw and bw > 0.0 else 0.0
        return lnet, bw, cost, lnet + xfer

    # ---------- arrivals / routing ----------
    def _on_arrival(self, jtype: str, ing_name: str):
        jid = next(self._jid)
        size = sample_job_size(jtype, self.rng)
        _rec = None
        if self.arrival_recorder is not None:
            _rec = [self.now, ing_name, jtype, size, None]
            self.arrival_recorder.append(_rec)
        job = JobState(jid=jid, ingress=ing_name, jtype=jtype, size=size,
                       arrival_time=self.now)

        if self.algo == "eco_route":
            best = None
            for dc in self.dcs.values():
                lnet, bw, cost, transfer_s = self._net_tuple(ing_name, dc.name, job)
                score, n_star, f_star = self._score_dc(dc, job)
                cand = (score, dc.name, lnet, bw, cost, transfer_s, n_star, f_star)
                if best is None or cand[0] < best[0]:
                    best = cand
            _, dc_name, lnet, bw, cost, transfer_s, n_star, f_star = best
            job.eco_hint = (n_star, f_star)
        elif self.algo == "chsac_af" and self.rl is not None:
            obs = self._rl_obs()
            m_dc, m_g = self._rl_masks()
            a = self.rl.select_action(obs, m_dc, m_g, deterministic=self.rl_deterministic)
            dc_name = self._dc_names[int(a["dc"])]
            n_sel = int(a["g"]) + 1
            job.rl_state0 = obs
            job.rl_action = {"dc_idx": int(a["dc"]), "g_idx": int(a["g"]), "n": n_sel}
            lnet, bw, cost, transfer_s = self._net_tuple(ing_name, dc_name, job)
        elif self.sc.router.use_weighted:
            dc_name = self._weighted_route(ing_name, job)
            lnet, bw, cost, transfer_s = self._net_tuple(ing_name, dc_name, job)
        else:
            dc_name = self.rng.choice(self._dc_names)
            lnet, bw, cost, transfer_s = self._net_tuple(ing_name, dc_name, job)

        if _rec is not None:
            _rec[4] = dc_name
        self._schedule(self.now + transfer_s, "xfer_done", {
            "ing": ing_name, "dc": dc_name, "jid": jid, "job": job,
            "net_lat_s": lnet, "net_bw_gbps": bw, "net_path_cost_gb": cost})

        # self-exciting chain: schedule this ingress's next arrival
        arr = self.arr_inf if jtype == "inference" else self.arr_trn
        self._schedule(self.now + arr.next_interarrival(self.now, self.rng),
                       "arrival_inf" if jtype == "inference" else "arrival_trn",
                       {"ing": ing_name})

    def _weighted_route(self, ing_name: str, job: JobState) -> str:
        """Weighted router score  wE*E1 + wL*Lnet + wC*(E1*CI)  with optional
        power-of-d candidate sampling.  The reference defines these weights
        (router.py:3-9) but routes uniformly at random; this is the opt-in
        implementation (Scenario.router.use_weighted=True) — SURVEY §2 row 10
        'implement the weighted score for parity-plus'."""
        rp = self.sc.router
        if rp.d_choices and 0 < rp.d_choices < len(self._dc_names):
            cands = self.rng.sample(self._dc_names, rp.d_choices)
        else:
            cands = self._dc_names
        best_name, best_score = None, None
        for name in cands:
            dc = self.dcs[name]
            pC, tC = self._coeffs(name, job.jtype)
            _, _, _, _, E_unit = best_nf_grid(
                self.sc.policy.max_gpus_per_job, dc.freq_levels, pC, tC,
                objective="energy", deadline_s=job.deadline)
            E1 = E_unit * job.size  # J per job at the energy-optimal point
            lnet, _, _, _ = self._net_tuple(ing_name, name, job)
            CI = self.sc.carbon_intensity.get(name, 0.0)
            score = rp.w_energy * E1 + rp.w_latency * lnet + rp.w_carbon * (E1 * CI)
            if best_score is None or score < best_score:
                best_name, best_score = name, score
        return best_name

    def _score_dc(self, dc: DataCenterState, job: JobState):
        """Eco-route DC score: (score, n*, f*) for the configured objective
        (reference _score_dc_for_job, :1007-1039)."""
        pC, tC = self._coeffs(dc.name, job.jtype)
        obj = self.eco_objective
        ddl = job.deadline
        CI = self.sc.carbon_intensity.get(dc.name, 0.0)
        price = self._price_kwh()
        if obj == "carbon":
            n, f, T, P, E = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                         pC, tC, objective="carbon",
                                         carbon_intensity=CI, deadline_s=ddl)
            score = (E * job.size) / J_PER_KWH * CI
        elif obj == "cost":
            n, f, T, P, E = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                         pC, tC, objective="cost",
                                         price_kwh=price, deadline_s=ddl)
            score = (E * job.size) / J_PER_KWH * price
        else:
            n, f, T, P, E = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                         pC, tC, objective="energy", deadline_s=ddl)
            score = E * job.size
        return score, n, f

    def _price_kwh(self) -> float:
        hour = int((self.now % 86400) // 3600)
        return float(self.sc.energy_price_hourly.get(hour, 0.0))

    # ---------- DC-side scheduling ----------
    def _on_transfer_done(self, payload: dict):
        dc = self.dcs[payload["dc"]]
        job: JobState = payload["job"]
        job.dc_name = dc.name
        job.arrival_time = self.now       # arrival at the DC
        job.net_latency_s = payload["net_lat_s"]
        if dc.free_gpus > 0:
            started = self._decide_and_start(dc, job)
            if started:
                return
        (dc.q_inf if job.jtype == "inference" else dc.q_train).append(job)

    def _decide_and_start(self, dc: DataCenterState, job: JobState) -> bool:
        """Per-algorithm (n, f) decision + job start for a job admitted to a DC
        with free GPUs (reference _handle_transfer_done dispatch, :602-676).
        Returns True if the job was started (or queue-routed internally)."""
        algo = self.algo
        if algo == "joint_nf":
            pC, tC = self._coeffs(dc.name, job.jtype)
            n, f, *_ = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                    pC, tC, objective="energy", carbon_intensity=0.0,
                                    deadline_s=job.deadline)
            self._start_with_nf(dc, job, n, f)
            return True
        if algo == "bandit":
            n = min(dc.free_gpus, self.sc.policy.max_gpus_per_job)
            f = self.bandit.select(dc.name, job.jtype, dc.freq_levels)
            self._start_with_nf(dc, job, n, f)
            return True
        if algo == "carbon_cost":
            pC, tC = self._coeffs(dc.name, job.jtype)
            price = self._price_kwh()
            CI = self.sc.carbon_intensity.get(dc.name, 0.0)
            if price > 0.0:
                n, f, *_ = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                        pC, tC, objective="cost", price_kwh=price,
                                        deadline_s=job.deadline)
            else:
                n, f, *_ = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                        pC, tC, objective="carbon", carbon_intensity=CI,
                                        deadline_s=job.deadline)
            self._start_with_nf(dc, job, n, f)
            return True
        if algo == "chsac_af" and getattr(job, "rl_action", None) is not None:
            a = job.rl_action
            n = max(1, min(a["n"], dc.free_gpus, self.sc.policy.max_gpus_per_job))
            f = self._energy_freq_with_deadline(dc, job, n)
            self._start_with_nf(dc, job, n, float(f))
            return True
        if algo == "debug":
            pC, tC = self._coeffs(dc.name, job.jtype)
            n = self.num_fixed_gpus
            f = self.fixed_freq if self.fixed_freq else best_energy_freq(n, dc.freq_levels, pC, tC)
            self._start_with_nf(dc, job, n, f)
            return True
        # heuristic path (default_policy, cap_*, eco_route)
        g = heuristic_allocate(dc, job, self.sc.policy)
        if g > 0:
            self._start_heuristic(dc, job, g)
            return True
        return False

    def _energy_freq_with_deadline(self, dc: DataCenterState, job: JobState, n: int) -> float:
        """Energy-optimal f at fixed n, raised to the smallest ladder step that
        meets the job deadline if one exists (reference :651-667)."""
        pC, tC = self._coeffs(dc.name, job.jtype)
        levels = sorted(dc.freq_levels) if dc.freq_levels else [dc.current_freq]
        f_opt = best_energy_freq(n, levels, pC, tC)
        ddl = job.deadline
        if ddl is not None:
            if job.size * unit_time_s(n, f_opt, tC) > ddl:
                for f_cand in levels:
                    if job.size * unit_time_s(n, f_cand, tC) <= ddl:
                        f_opt = f_cand
                        break
                else:
                    f_opt = levels[-1]
        return f_opt

    def _start_heuristic(self, dc: DataCenterState, job: JobState, gpus: int):
        """Start at DC-level frequency (reference _start_job, :680-699)."""
        if gpus <= 0:
            (dc.q_inf if job.jtype == "inference" else dc.q_train).append(job)
            return
        dc.busy_gpus += gpus
        dc.running_jobs[job.jid] = (job, gpus)
        job.gpus_assigned = gpus
        job.start_time = self.now
        job.f_used = dc.current_freq
        job.units_total = job.size
        job.units_done = 0.0
        job.last_update = self.now
        job.ev_gen += 1
        _, tC = self._coeffs(dc.name, job.jtype)
        T = unit_time_s(gpus, dc.current_freq, tC)
        self._schedule(self.now + job.size * T, "job_finish",
                       {"dc": dc.name, "jid": job.jid, "gen": job.ev_gen})

    def _start_with_nf(self, dc: DataCenterState, job: JobState, n: int, f: float):
        """Start with explicit per-job (n, f) DVFS (reference _start_job_with_nf,
        :960-980); n is clamped to free GPUs."""
        n = max(1, min(n, dc.free_gpus))
        if n <= 0:
            (dc.q_inf if job.jtype == "inference" else dc.q_train).append(job)
            return
        dc.busy_gpus += n
        dc.running_jobs[job.jid] = (job, n)
        job.gpus_assigned = n
        job.start_time = self.now
        job.f_used = f
        job.units_total = job.size
        job.units_done = 0.0
        job.last_update = self.now
        job.ev_gen += 1
        _, tC = self._coeffs(dc.name, job.jtype)
        T = unit_time_s(n, f, tC)
        self._schedule(self.now + job.size * T, "job_finish",
                       {"dc": dc.name, "jid": job.jid, "gen": job.ev_gen})

    # ---------- progress / rescheduling / preemption ----------
    def _job_rate(self, dc, job, gpus, f):
        _, tC = self._coeffs(dc.name, job.jtype)
        return 1.0 / max(unit_time_s(gpus, f, tC), 1e-9)

    def _advance_progress(self, dc, job, gpus):
        rate = self._job_rate(dc, job, gpus, job.f_used or dc.current_freq)
        dt = max(0.0, self.now - job.last_update)
        job.units_done = min(job.units_total, job.units_done + rate * dt)
        job.last_update = self.now

    def _reschedule_job(self, dc, job, gpus, new_f):
        self._advance_progress(dc, job, gpus)
        job.f_used = new_f
        units_left = max(0.0, job.units_total - job.units_done)
        finish_in = units_left / max(self._job_rate(dc, job, gpus, new_f), 1e-9)
        job.ev_gen += 1
        self._schedule(self.now + finish_in, "job_finish",
                       {"dc": dc.name, "jid": job.jid, "gen": job.ev_gen})

    def _preempt_job(self, dc: DataCenterState, job: JobState, reason: str):
        if job.jid not in dc.running_jobs:
            return
        job, gpus = dc.running_jobs.pop(job.jid)
        self._advance_progress(dc, job, gpus)
        dc.preempted_jobs.append(PreemptedJobState(
            job=job, preempt_time=self.now, reason=reason,
            ckpt={"units_done": job.units_done, "f_used": job.f_used,
                  "gpus_assigned": gpus}))
        dc.busy_gpus -= gpus
        job.preempt_count += 1

    def _resume_preempted(self, dc: DataCenterState, pre: PreemptedJobState,
                          n_resume: int, f_resume: float) -> bool:
        job = pre.job
        if dc.free_gpus < n_resume:
            return False
        job.units_done = pre.ckpt["units_done"]
        job.f_used = f_resume
        job.gpus_assigned = n_resume
        job.last_update = self.now
        job.total_preempt_time += self.now - pre.preempt_time
        dc.busy_gpus += n_resume
        dc.running_jobs[job.jid] = (job, n_resume)
        units_left = max(0.0, job.units_total - job.units_done)
        _, tC = self._coeffs(dc.name, job.jtype)
        T = unit_time_s(n_resume, f_resume, tC)
        job.ev_gen += 1
        self._schedule(self.now + units_left / max(1.0 / T, 1e-9), "job_finish",
                       {"dc": dc.name, "jid": job.jid, "gen": job.ev_gen})
        dc.preempted_jobs.remove(pre)
        self.logger.info(f"Resume job {job.jid} at {self._hour()}.")
        return True

    def _hour(self) -> int:
        return int((self.now % 86400) // 3600)

    # ---------- job completion ----------
    def _on_job_finish(self, dc_name: str, jid: int):
        dc = self.dcs[dc_name]
        tup = dc.running_jobs.pop(jid, None)
        if not tup:
            return
        job, g = tup
        dc.busy_gpus = max(0, dc.busy_gpus - g)
        job.finish_time = self.now
        self.jobs_completed += 1

        # credit the remainder job-units since the last log tick; the remainder
        # window is finish_time mod log_interval (reference :711, quirk kept).
        self._accumulate_job_unit(dc, job, job.finish_time % self.log_interval)

        pC, tC = self._coeffs(dc.name, job.jtype)
        f_used = job.f_used
        T_pred, P_pred, E_pred = energy_tuple(g, f_used, pC, tC)

        # RL metrics (sliding p99 windows etc., reference :718-754)
        E_job_kwh = float(E_pred) * float(job.size) / J_PER_KWH
        sojourn_s = max(0.0, job.finish_time - job.start_time)
        buf = self._lat_hist.setdefault(job.jtype, [])
        buf.append(sojourn_s)
        if len(buf) > 2048:
            del buf[: len(buf) - 2048]
        mean_ms = (sum(buf) / len(buf)) * 1000.0 if buf else sojourn_s * 1000.0
        p99_ms = self._p99_ms(buf, sojourn_s)
        P_now = self._dc_power_w(dc)
        rl_metrics = {"energy_kwh": E_job_kwh, "units_processed": float(job.size),
                      "mean_latency_ms": mean_ms, "p99_latency_ms": p99_ms,
                      "power_W": float(P_now), "power_state_changes": 0}
        self.logger.debug({k: round(v, 4) if isinstance(v, (int, float)) else v
                           for k, v in rl_metrics.items()})

        if (self.algo == "chsac_af" and self.rl is not None
                and getattr(job, "rl_action", None) is not None
                and getattr(job, "rl_state0", None) is not None):
            self._rl_on_finish(dc, job, g, f_used, rl_metrics)

        if self.job_writer is not None:
            self.job_writer.row(job.jid, job.ingress, job.jtype, job.size, dc.name,
                                f_used, g, job.net_latency_s, job.start_time,
                                job.finish_time, job.preempt_count,
                                T_pred, P_pred, E_pred)

        if self.bandit is not None:
            self.bandit.update(dc.name, job.jtype, f_used, E_pred)

        # elastic reallocation (chsac_af only; reference :829-837)
        if self.elastic_scaling and job.jtype == "training":
            n_train = sum(1 for j, _ in dc.running_jobs.values() if j.jtype == "training")
            if n_train > 1:
                pre_list = self._preempt_all_training(dc, "Re-allocate on job completion.")
                self.logger.info(f"Preempt all training jobs of {dc.name} at "
                                 f"{self._hour()} upon job completion.")
                if pre_list:
                    self._rl_reallocate(dc, pre_list)

        self._drain_queues(dc)

    def _p99_ms(self, buf, sojourn_s):
        if len(buf) >= 5:
            import numpy as np
            return float(np.percentile(buf, 99) * 1000.0)
        return sojourn_s * 1000.0

    def _accumulate_job_unit(self, dc: DataCenterState, job: JobState, window_s: float):
        """acc += time-window * 1/T(n, f_used) (reference :951-958)."""
        _, tC = self._coeffs(dc.name, job.jtype)
        tpt = 1.0 / unit_time_s(job.gpus_assigned, job.f_used, tC)
        dc.accumulated_job_unit += tpt * window_s

    def _preempt_all_training(self, dc: DataCenterState, reason: str):
        out = []
        to_preempt = [(jid, job, gp) for jid, (job, gp) in list(dc.running_jobs.items())
                      if job.jtype == "training"]
        for jid, job, gp in to_preempt:
            self._preempt_job(dc, job, reason)
            pre = next((p for p in dc.preempted_jobs if p.job.jid == jid), None)
            if pre:
                out.append(pre)
        return out

    def _rl_reallocate(self, dc: DataCenterState, pre_list):
        """Ask the agent for a new n per preempted training job; f is
        energy-optimal at that n (reference _rl_reallocate_training_jobs,
        :498-534).  Unlike the reference, a failed resume (not enough free
        GPUs) re-queues the checkpointed job on the training queue instead of
        stranding it (fix of SURVEY Appendix A.6, documented)."""
        for pre in pre_list:
            job = pre.job
            obs = self._rl_obs()
            m_dc, m_g = self._rl_masks()
            a = self.rl.select_action(obs, m_dc, m_g, deterministic=self.rl_deterministic)
            n_rl = max(1, min(int(a["g"]) + 1, dc.free_gpus,
                              self.sc.policy.max_gpus_per_job))
            f_rl = self._energy_freq_with_deadline(dc, job, n_rl)
            if not self._resume_preempted(dc, pre, n_rl, f_rl):
                dc.preempted_jobs.remove(pre)
                job.units_done = pre.ckpt["units_done"]
                dc.q_train.append(job)
                self.logger.info(f"Resume of job {job.jid} deferred (no free GPUs); re-queued.")

    def _drain_queues(self, dc: DataCenterState):
        """Start queued jobs while GPUs are free, inference first
        (reference :839-927)."""
        while dc.free_gpus > 0:
            nxt = None
            if self.sc.policy.inf_priority and dc.q_inf:
                nxt = dc.q_inf.pop(0)
            elif dc.q_train:
                nxt = dc.q_train.pop(0)
            if nxt is None:
                break

            if self.algo == "chsac_af" and self.rl is not None:
                obs = self._rl_obs()
                m_dc, m_g = self._rl_masks()
                a = self.rl.select_action(obs, m_dc, m_g, deterministic=self.rl_deterministic)
                dc_tgt = self.dcs[self._dc_names[int(a["dc"])]]
                if dc_tgt.free_gpus <= 0:
                    (dc.q_inf if nxt.jtype == "inference" else dc.q_train).insert(0, nxt)
                    break
                n_sel = max(1, min(int(a["g"]) + 1, dc_tgt.free_gpus,
                                   self.sc.policy.max_gpus_per_job))
                f_sel = float(self._energy_freq_with_deadline(dc_tgt, nxt, n_sel))
                self._start_with_nf(dc_tgt, nxt, n_sel, f_sel)
                nxt.rl_state0 = obs
                nxt.rl_action = {"dc_idx": int(a["dc"]), "g_idx": int(a["g"]), "n": n_sel}
                break  # DC state changed; at most one drain per finish (:890)

            if self.algo == "joint_nf":
                pC, tC = self._coeffs(dc.name, nxt.jtype)
                n, f, *_ = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                        pC, tC, objective="energy", carbon_intensity=0.0,
                                        deadline_s=nxt.deadline)
                self._start_with_nf(dc, nxt, n, f)
            elif self.algo == "bandit":
                n = min(dc.free_gpus, self.sc.policy.max_gpus_per_job)
                f = self.bandit.select(dc.name, nxt.jtype, dc.freq_levels)
                self._start_with_nf(dc, nxt, n, f)
            elif self.algo == "carbon_cost":
                pC, tC = self._coeffs(dc.name, nxt.jtype)
                CI = self.sc.carbon_intensity.get(dc.name, 0.0)
                n, f, *_ = best_nf_grid(self.sc.policy.max_gpus_per_job, dc.freq_levels,
                                        pC, tC, objective="carbon", carbon_intensity=CI,
                                        deadline_s=nxt.deadline)
                self._start_with_nf(dc, nxt, n, f)
            else:
                g = heuristic_allocate(dc, nxt, self.sc.policy)
                if g <= 0:
                    (dc.q_inf if nxt.jtype == "inference" else dc.q_train).insert(0, nxt)
                    break
                self._start_heuristic(dc, nxt, g)

    # ---------- logging tick + power-cap control ----------
    def _on_log(self, interval: float):
        for name, dc in self.dcs.items():
            run_total = len(dc.running_jobs)
            run_inf = sum(1 for j, _ in dc.running_jobs.values() if j.jtype == "inference")
            run_trn = run_total - run_inf
            util_inst = (dc.busy_gpus / dc.total_gpus) if dc.total_gpus else 0.0
            elapsed = max(1e-9, self.now - (dc.util_begin_ts or self.now))
            util_avg = (dc.util_gpu_time / (dc.total_gpus * elapsed)) if dc.total_gpus else 0.0
            power_now = self._dc_power_w(dc)
            for job, _ in dc.running_jobs.values():
                self._accumulate_job_unit(dc, job, interval)
            if self.cluster_writer is not None:
                self.cluster_writer.row(self.now, name, dc.current_freq,
                                        dc.busy_gpus, dc.free_gpus,
                                        run_total, run_inf, run_trn,
                                        len(dc.q_inf), len(dc.q_train),
                                        util_inst, util_avg, dc.accumulated_job_unit,
                                        power_now, dc.energy_joules)
        self._schedule(self.now + interval, "log", {"interval": interval})

    def _control(self):
        """Power-cap controller, fired on every log tick (reference _control,
        :207-315 — note it fires at log_interval, not control_interval;
        SURVEY Appendix A.5, kept)."""
        if self.power_cap <= 0:
            return
        if self.algo not in ("cap_uniform", "cap_greedy"):
            if self.algo in ("eco_route", "carbon_cost"):
                for dc in self.dcs.values():
                    if dc.busy_gpus == 0 and dc.freq_levels:
                        dc.current_freq = min(dc.freq_levels)
            return

        totalP = sum(self._dc_power_w(dc) for dc in self.dcs.values())
        if totalP <= self.power_cap - self.cap_margin:
            return
        deficit = max(0.0, totalP - self.power_cap)
        if deficit <= 1e-6:
            return
        if self.algo == "cap_uniform":
            return self._cap_uniform(deficit)
        self._cap_greedy(deficit, totalP)

    def _cap_uniform(self, deficit: float):
        """DC-level discrete down-steps, biggest delta-P first.  Under per-job
        DVFS the power probe is insensitive to the DC-level frequency, so every
        delta is 0 and this exits immediately — reference behaviour preserved
        bit-for-bit (reference :181-205; see module docstring)."""
        guard = 10000
        while deficit > 1e-6 and guard > 0:
            guard -= 1
            best_dc, best_dp, best_f = None, 0.0, None
            for dc in self.dcs.values():
                levels = dc.freq_levels
                try:
                    idx = levels.index(dc.current_freq)
                except ValueError:
                    idx = min(range(len(levels)), key=lambda i: abs(levels[i] - dc.current_freq))
                if idx == 0:
                    continue
                dp = self._dc_power_w(dc) - self._dc_power_w(dc)  # f_used-based: always 0
                if dp > best_dp + 1e-9:
                    best_dc, best_dp, best_f = dc, dp, levels[idx - 1]
            if not best_dc or best_dp <= 1e-9:
                break
            best_dc.current_freq = best_f
            deficit -= best_dp

    def _cap_greedy(self, deficit: float, totalP: float):
        """Per-job atom-based capping: apply down-atoms cheapest-rho-first with
        exact power re-estimation after each (reference :248-315)."""
        guard = 10000
        while deficit > 1e-6 and guard > 0:
            guard -= 1
            tasks = []
            for dc in self.dcs.values():
                levels = dc.freq_levels or []
                if not levels:
                    continue
                f_min = min(levels)
                for job, g in list(dc.running_jobs.values()):
                    cur_f = job.f_used or dc.current_freq
                    if cur_f <= f_min + 1e-12:
                        continue
                    pc, tc = self._coeff_cache[(dc.name, job.jtype)]
                    tasks.append(RunningTask(
                        job_id=job.jid, dc_name=dc.name, n=g, f=cur_f,
                        freq_levels=levels, pc=pc, tc=tc))
            if not tasks:
                break
            _, down_atoms = aggregate_atoms(tasks)
            if not down_atoms:
                break
            applied_any = False
            for atom in down_atoms:
                if deficit <= 1e-6:
                    break
                dc = self.dcs.get(atom.dc_name)
                if dc is None:
                    continue
                tup = dc.running_jobs.get(atom.job_id)
                if not tup:
                    continue
                job, g = tup
                cur_f = job.f_used or dc.current_freq
                if atom.f_to >= cur_f - 1e-12:
                    continue
                self._reschedule_job(dc, job, g, atom.f_to)
                applied_any = True
                totalP = sum(self._dc_power_w(d_) for d_ in self.dcs.values())
                deficit = max(0.0, totalP - self.power_cap)
                if deficit <= 1e-6:
                    break
            if not applied_any:
                break

    # ---------- RL glue ----------
    def _rl_obs(self):
        """[now] + per-DC [total, busy, free, current_f, qlen_inf, qlen_trn]
        (reference _upgr_obs, :1041-1053)."""
        import numpy as np
        feats = []
        for dc in self.dcs.values():
            total = float(dc.total_gpus)
            busy = float(dc.busy_gpus)
            feats.extend([total, busy, max(0.0, total - busy),
                          float(dc.current_freq), float(len(dc.q_inf)),
                          float(len(dc.q_train))])
        return np.asarray([float(self.now)] + feats, dtype=np.float32)

    def _rl_masks(self):
        """DC mask (has free GPUs) and g mask (1..N <= max free); the g mask is
        capped at 1 GPU when recent p99 is comfortably inside SLA
        (reference _upgr_masks, :1055-1082)."""
        import numpy as np
        dc_mask, max_free = [], 0
        for dc in self.dcs.values():
            free = max(0, dc.total_gpus - int(dc.busy_gpus))
            dc_mask.append(free > 0)
            max_free = max(max_free, free)
        n_choices = int(self.sc.policy.max_gpus_per_job)
        g_mask = [(i + 1) <= max_free for i in range(n_choices)]
        buf = self._lat_hist.get("training") or self._lat_hist.get("inference")
        if buf and len(buf) >= 5 and self.rl is not None:
            import numpy as np2
            p99_recent_ms = float(np2.percentile(buf, 99) * 1000.0)
            target = self.rl.constraint_target("latency_p99")
            if target is not None and p99_recent_ms < 0.9 * target:
                cap = 1
                g_mask = [(i + 1) <= min(cap, max_free) for i in range(n_choices)]
        return (np.asarray(dc_mask, bool), np.asarray(g_mask, bool))

    def _min_n_for_sla(self, dc, job, f, sla_ms):
        _, tC = self._coeffs(dc.name, job.jtype)
        for n_try in range(1, self.sc.policy.max_gpus_per_job + 1):
            if job.size * unit_time_s(n_try, f, tC) * 1000.0 <= sla_ms:
                return n_try
        return self.sc.policy.max_gpus_per_job

    def _rl_on_finish(self, dc, job, g, f_used, rl_metrics):
        """Reward shaping, transition push, SAC train step (reference :757-811)."""
        E_unit_kwh = float(rl_metrics["energy_kwh"]) / (float(rl_metrics["units_processed"]) + 1e-9)
        n = max(1, int(job.rl_action["n"]))
        r = -E_unit_kwh + 0.05 * (1.0 / n)
        sla_target = self.rl.constraint_target("latency_p99") or self.sla_p99_ms
        n_min = self._min_n_for_sla(dc, job, f_used, sla_target)
        costs = {"latency_p99": float(rl_metrics["p99_latency_ms"]),
                 "power": float(rl_metrics["power_W"]),
                 "gpu_over": float(max(0, g - n_min))}
        m_dc, m_g = self._rl_masks()
        self.replay.add(s=job.rl_state0, s_next=self._rl_obs(),
                        a_dc=int(job.rl_action["dc_idx"]), a_g=int(job.rl_action["g_idx"]),
                        r=r, costs=costs, done=True, mask_dc=m_dc, mask_g=m_g)
        if self.replay.size >= self._rl_warmup:
            batch = self.replay.sample(self._rl_batch)
            stats = self.rl.train_step(batch)
            self.rl_updates += 1
            if stats:
                self.logger.info({k: round(v, 4) if isinstance(v, (int, float)) else v
                                  for k, v in stats.items()})
        job.rl_state0 = None
        job.rl_action = None
