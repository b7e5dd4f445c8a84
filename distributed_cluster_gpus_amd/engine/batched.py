"""BatchedEngine: MI355X Monte-Carlo replica engine orchestrator.

Allocates the replica-major SoA device state (models/scenario tables +
per-replica simulation state) as torch tensors on the GPU, then drives the
gfx950 advance kernel (ops/csrc/hip/replica_engine.hip) in chunks until every
replica reaches end_time.  One wavefront advances one replica; R replicas run
concurrently.  Memory is sized for 288 GB HBM3E — the default 4096-replica
paper-config state is ~2 GB, so tens of thousands of replicas per GPU fit
comfortably (BASELINE.json config 3: 65k replicas over 8 GPUs).

Multi-GPU: replicas are sharded across ranks (parallel/sharding.py); RNG
streams key on GLOBAL replica ids so results are independent of world size;
metric reduction is one small RCCL all-reduce at the end.

Logging: a designated replica (global replica 0, on whichever rank owns it)
records cluster/job rows in device buffers; the host formats them through the
same CSV writers as the scalar engines, so the batched path emits the exact
log schema.  Other replicas contribute to Monte-Carlo aggregate metrics only
— a capability the scalar reference cannot express (SURVEY §6 north-star).

GPU-required: this engine fails loudly if CUDA/ROCm or the _sim_hip extension
is unavailable (no silent CPU fallback).
"""
import math
import os
from typing import Optional

import numpy as np
import torch

from ..models.arrivals import ArrivalProcess
from ..models.scenario import PAYLOAD_GB, Scenario
from ..ops import load_sim_hip
from ..utils.csvlog import ClusterLogWriter, JobLogWriter
from ..utils.timers import ThroughputMeter
from .oracle import ALGOS

_ALGO_IDS = {"default_policy": 0, "cap_uniform": 1, "cap_greedy": 2,
             "joint_nf": 3, "bandit": 4, "carbon_cost": 5, "eco_route": 6,
             "debug": 7}
_ECO_IDS = {"energy": 0, "carbon": 1, "cost": 2}
INF = 1e300


class BatchedEngine:
    def __init__(self, scenario: Scenario,
                 arrival_inf: ArrivalProcess, arrival_trn: ArrivalProcess,
                 *, algo: str = "default_policy", replicas: int = 4096,
                 duration: float = 3600.0, log_interval: float = 10.0,
                 out_dir: Optional[str] = None, seed: int = 42,
                 power_cap: float = 0.0, control_interval: float = 5.0,
                 elastic_scaling: bool = False, eco_objective: str = "energy",
                 num_fixed_gpus: int = 1, fixed_freq: Optional[float] = None,
                 logger=None, show_progress: bool = False,
                 device: Optional[torch.device] = None,
                 rank: int = 0, world: int = 1,
                 tcap: int = 256, qcap: int = 24576,
                 events_per_launch: int = 50000,
                 enable_logs: bool = True, **_unused_rl_kwargs):
        if algo not in ALGOS:
            raise ValueError(f"unknown algo {algo!r}")
        if algo == "chsac_af":
            raise NotImplementedError(
                "chsac_af on the batched engine lands with the RL-batch phase; "
                "use --engine oracle for RL runs this round")
        if not torch.cuda.is_available():
            raise RuntimeError("BatchedEngine requires a ROCm GPU "
                               "(no silent CPU fallback)")
        self._mod = load_sim_hip()  # raises if the gfx950 extension is missing
        self.sc = scenario
        self.algo = algo
        self.device = device or torch.device("cuda", torch.cuda.current_device())
        self.end_time = float(duration)
        self.log_interval = float(log_interval)
        self.events_per_launch = int(events_per_launch)
        self.out_dir = out_dir
        self.rl = None

        from ..parallel.sharding import replica_shard
        shard = replica_shard(int(replicas), rank, world)
        self.shard = shard
        R = shard.count
        self.R = R
        n_dc, n_ing, n_freq = scenario.n_dc, scenario.n_ing, scenario.n_freq
        total_slots = int(scenario.total_gpus.sum())
        slot_off = np.zeros(n_dc + 1, np.int32)
        slot_off[1:] = np.cumsum(scenario.total_gpus)
        slot_dc = np.zeros(total_slots, np.int32)
        for d in range(n_dc):
            slot_dc[slot_off[d]:slot_off[d + 1]] = d

        dev = self.device
        f64 = dict(dtype=torch.float64, device=dev)
        f32 = dict(dtype=torch.float32, device=dev)
        i64 = dict(dtype=torch.int64, device=dev)
        i32 = dict(dtype=torch.int32, device=dev)
        i16 = dict(dtype=torch.int16, device=dev)
        i8 = dict(dtype=torch.int8, device=dev)

        def T(arr, **kw):
            return torch.as_tensor(np.ascontiguousarray(arr), **kw).to(dev)

        t = {}
        # scenario constants
        t["freq_levels"] = T(scenario.freq_levels, dtype=torch.float64)
        t["pc"] = T(scenario.power_coeffs.reshape(-1), dtype=torch.float64)
        t["lc"] = T(scenario.latency_coeffs.reshape(-1), dtype=torch.float64)
        t["wan_lat"] = T(np.asarray(scenario.wan_latency_s).reshape(-1), dtype=torch.float64)
        t["wan_bw"] = T(np.asarray(scenario.wan_bottleneck_gbps).reshape(-1), dtype=torch.float64)
        t["carbon"] = T(scenario.carbon_vec(), dtype=torch.float64)
        t["price24"] = T(scenario.price_vec24(), dtype=torch.float64)
        t["total_gpus"] = T(scenario.total_gpus, dtype=torch.int32)
        t["p_idle"] = T(scenario.p_idle, dtype=torch.float64)
        t["p_sleep"] = T(scenario.p_sleep, dtype=torch.float64)
        t["p_peak"] = T([scenario.gpu_specs[n].p_peak for n in scenario.dc_names],
                        dtype=torch.float64)
        t["pow_alpha"] = T([scenario.gpu_specs[n].alpha for n in scenario.dc_names],
                           dtype=torch.float64)
        t["power_gating"] = T(scenario.power_gating.astype(np.int32), dtype=torch.int32)
        t["default_freq"] = T(scenario.default_freq, dtype=torch.float64)
        t["slot_off"] = T(slot_off, dtype=torch.int32)
        t["slot_dc"] = T(slot_dc, dtype=torch.int32)
        # per-replica state
        t["now"] = torch.full((R,), -1.0, **f64)
        t["next_log"] = torch.full((R,), self.log_interval, **f64)
        t["rng_ctr"] = torch.zeros(R, **i64)
        t["jid_ctr"] = torch.zeros(R, **i32)
        t["done"] = torch.zeros(R, **i32)
        t["err"] = torch.zeros(R, **i32)
        t["arr_next"] = torch.full((R, n_ing * 2), INF, **f64)
        t["busy"] = torch.zeros((R, n_dc), **i32)
        t["cur_freq"] = torch.empty((R, n_dc), **f32)
        t["cur_freq"][:] = torch.as_tensor(scenario.default_freq, dtype=torch.float32,
                                           device=dev)
        t["energy_j"] = torch.zeros((R, n_dc), **f64)
        t["util_time"] = torch.zeros((R, n_dc), **f64)
        t["util_begin"] = torch.full((R, n_dc), -1.0, **f64)
        t["acc_unit"] = torch.zeros((R, n_dc), **f64)
        t["p_active"] = torch.zeros((R, n_dc), **f64)
        t["sum_tpt"] = torch.zeros((R, n_dc), **f64)
        t["n_running"] = torch.zeros((R, n_dc), **i32)
        t["dc_min_finish"] = torch.full((R, n_dc), INF, **f64)
        t["dc_min_slot"] = torch.full((R, n_dc), -1, **i32)
        t["s_finish"] = torch.full((R, total_slots), INF, **f64)
        t["s_start"] = torch.zeros((R, total_slots), **f64)
        t["s_size"] = torch.zeros((R, total_slots), **f32)
        t["s_fused"] = torch.zeros((R, total_slots), **f32)
        t["s_netlat"] = torch.zeros((R, total_slots), **f32)
        t["s_jid"] = torch.zeros((R, total_slots), **i32)
        t["s_gpus"] = torch.zeros((R, total_slots), **i16)
        t["s_jtype"] = torch.zeros((R, total_slots), **i8)
        t["s_ing"] = torch.zeros((R, total_slots), **i8)
        t["x_time"] = torch.full((R, tcap), INF, **f64)
        t["x_size"] = torch.zeros((R, tcap), **f32)
        t["x_netlat"] = torch.zeros((R, tcap), **f32)
        t["x_jid"] = torch.zeros((R, tcap), **i32)
        t["x_dc"] = torch.zeros((R, tcap), **i8)
        t["x_jtype"] = torch.zeros((R, tcap), **i8)
        t["x_ing"] = torch.zeros((R, tcap), **i8)
        t["q_head"] = torch.zeros((R, n_dc, 2), **i32)
        t["q_len"] = torch.zeros((R, n_dc, 2), **i32)
        t["q_size"] = torch.zeros((R, n_dc, 2, qcap), **f32)
        # queue aux fields (net latency / jid / ingress) are only consumed by
        # the logging replica's job rows -> single-replica allocation
        t["q_netlat"] = torch.zeros((n_dc, 2, qcap), **f32)
        t["q_jid"] = torch.zeros((n_dc, 2, qcap), **i32)
        t["q_ing"] = torch.zeros((n_dc, 2, qcap), **i8)
        nb = 1 if algo != "bandit" else R
        t["b_n"] = torch.zeros((nb, n_dc, 2, n_freq), **i32)
        t["b_s"] = torch.zeros((nb, n_dc, 2, n_freq), **f32)
        t["b_t"] = torch.zeros(R, **i64)
        t["ev_count"] = torch.zeros(R, **i64)
        t["jobs_done"] = torch.zeros(R, **i64)
        t["jobs_done_inf"] = torch.zeros(R, **i64)
        t["sum_lat"] = torch.zeros(R, **f64)
        t["sum_lat_inf"] = torch.zeros(R, **f64)
        t["sum_wait"] = torch.zeros(R, **f64)

        # logging buffers (global replica 0 lives on rank 0 shard)
        self.log_replica = 0 if (enable_logs and shard.start == 0) else -1
        n_ticks = int(math.ceil(self.end_time / self.log_interval)) + 2
        cl_cap = (n_dc * n_ticks + 64) if self.log_replica >= 0 else 1
        jl_cap = 400_000 if self.log_replica >= 0 else 1
        t["cl_count"] = torch.zeros(1, **i32)
        t["cl_rows"] = torch.zeros((cl_cap, 16), **f64)
        t["jl_count"] = torch.zeros(1, **i32)
        t["jl_rows"] = torch.zeros((jl_cap, 10), **f64)

        # seed the initial arrival times on host (one inf + one trn per
        # ingress per replica), Philox-consistent with the device streams:
        # the kernel's first draws start at ctr = n_streams; host uses
        # ctr = stream index for the seed draws.
        arr_np = self._seed_arrivals(arrival_inf, arrival_trn, seed, shard)
        t["arr_next"].copy_(torch.as_tensor(arr_np, dtype=torch.float64))
        t["rng_ctr"].fill_(n_ing * 2)  # host consumed one block per stream

        self.t = t
        self.arrival_inf, self.arrival_trn = arrival_inf, arrival_trn

        cfg = {
            "n_rep": R, "n_dc": n_dc, "n_ing": n_ing, "n_freq": n_freq,
            "total_slots": total_slots, "tcap": tcap, "qcap": qcap,
            "end_time": self.end_time, "log_interval": self.log_interval,
            "algo": _ALGO_IDS[algo],
            "max_gpj": int(scenario.policy.max_gpus_per_job),
            "inf_priority": int(scenario.policy.inf_priority),
            "scale_out_low": int(scenario.policy.train_scale_out_low_freq),
            "energy_aware": int(scenario.policy.name == "energy_aware"),
            "dvfs_low": float(scenario.policy.dvfs_low),
            "dvfs_high": float(scenario.policy.dvfs_high),
            "power_cap": float(power_cap),
            "eco_obj": _ECO_IDS[eco_objective],
            "num_fixed": int(num_fixed_gpus),
            "fixed_freq": float(fixed_freq) if fixed_freq else 0.0,
            "payload_inf_gb": PAYLOAD_GB[0], "payload_trn_gb": PAYLOAD_GB[1],
            "arr_mode": [self._mode_id(arrival_inf.mode), self._mode_id(arrival_trn.mode)],
            "arr_rate": [float(arrival_inf.rate), float(arrival_trn.rate)],
            "arr_amp": [float(arrival_inf.amp), float(arrival_trn.amp)],
            "arr_period": [float(arrival_inf.period), float(arrival_trn.period)],
            "seed": int(seed), "rep_id_offset": int(shard.start),
            "log_replica": self.log_replica,
            "cl_cap": cl_cap, "jl_cap": jl_cap,
        }
        self._sim = self._mod.BatchedSimHip(t, cfg)
        self.meter = ThroughputMeter()

    @staticmethod
    def _mode_id(mode: str) -> int:
        return {"poisson": 0, "sinusoid": 1, "off": 2}[mode]

    def _seed_arrivals(self, arrival_inf, arrival_trn, seed, shard):
        """Host-side Philox draws for the initial inter-arrival per stream,
        matching the device recipe (philox.hpp)."""
        from ._philox_host import philox_u01, replica_key
        R = shard.count
        NS = self.sc.n_ing * 2
        out = np.full((R, NS), INF)
        for r in range(R):
            key = replica_key(seed, shard.start + r)
            ctr = 0
            for s in range(NS):
                jt = s & 1
                arr = arrival_inf if jt == 0 else arrival_trn
                # the kernel reserves one counter per stream for the seed draw;
                # thinning may need more draws -> sub-counter space: we use
                # counter = stream index, and for extra thinning draws we
                # borrow high bits (replica-unique; never reused by the device
                # which starts at ctr = NS).
                if arr.mode == "off" or arr.rate <= 0:
                    ctr += 1
                    continue
                if arr.mode == "poisson":
                    u = philox_u01(key, ctr)
                    out[r, s] = -math.log(1.0 - u) / arr.rate
                    ctr += 1
                else:  # sinusoid thinning at t=0
                    max_rate = arr.rate * (1.0 + abs(arr.amp))
                    sub = 0
                    ia = INF
                    while sub < 4096:
                        u1 = philox_u01(key, ctr + ((sub * 2 + 1) << 32))
                        w = -math.log(1.0 - u1) / max_rate
                        lam = max(0.0, arr.rate * (1.0 + arr.amp * math.sin(
                            2.0 * math.pi * (w % arr.period) / arr.period)))
                        u2 = philox_u01(key, ctr + ((sub * 2 + 2) << 32))
                        if u2 <= lam / max_rate:
                            ia = w
                            break
                        sub += 1
                    out[r, s] = ia
                    ctr += 1
        return out

    # ---------------- run ----------------
    def run(self):
        self.meter.start()
        t = self.t
        launches = 0
        while True:
            self._sim.advance(self.end_time, self.events_per_launch)
            torch.cuda.synchronize(self.device)
            launches += 1
            err = int(t["err"].max().item())
            if err != 0:
                raise RuntimeError(f"batched engine error flags: {err:#x} "
                                   f"(queue/transfer/slot/log overflow)")
            if bool(t["done"].min().item() == 1):
                break
            if launches > 100000:
                raise RuntimeError("batched engine failed to converge")
        self.meter.count = int(t["ev_count"].sum().item())
        self.meter.stop()
        if self.log_replica >= 0 and self.out_dir is not None:
            self._write_logs()
        return self.stats()

    def stats(self):
        t = self.t
        jobs = int(t["jobs_done"].sum().item())
        jobs_inf = int(t["jobs_done_inf"].sum().item())
        stats = {
            "events": int(t["ev_count"].sum().item()),
            "wall_s": self.meter.elapsed_s,
            "events_per_sec": self.meter.per_sec,
            "rl_updates": 0,
            "jobs_completed": jobs,
            "jobs_completed_inf": jobs_inf,
            "replicas": self.R,
            "total_energy_j": float(t["energy_j"].sum().item()),
            "mean_energy_j_per_replica": float(t["energy_j"].sum().item()) / max(1, self.R),
            "energy_j_replica_std": float(t["energy_j"].sum(dim=1).std().item()) if self.R > 1 else 0.0,
            "mean_latency_s": float(t["sum_lat"].sum().item()) / max(1, jobs),
            "mean_inf_latency_s": (float(t["sum_lat_inf"].sum().item()) / max(1, jobs_inf)),
            "launches": 0,
        }
        return stats

    def metrics_tensors(self):
        """Raw per-replica metric tensors (for cross-rank RCCL reductions)."""
        t = self.t
        return {k: t[k] for k in ("ev_count", "jobs_done", "jobs_done_inf",
                                  "sum_lat", "sum_lat_inf", "energy_j")}

    # ---------------- log formatting ----------------
    def _write_logs(self):
        t = self.t
        os.makedirs(self.out_dir, exist_ok=True)
        cw = ClusterLogWriter(os.path.join(self.out_dir, "cluster_log.csv"))
        n_cl = int(t["cl_count"].item())
        rows = t["cl_rows"][:n_cl].cpu().numpy()
        for row in rows:
            d = int(row[1])
            cw.row(row[0], self.sc.dc_names[d], row[2], int(row[3]), int(row[4]),
                   int(row[5]), int(row[6]), int(row[7]), int(row[8]), int(row[9]),
                   row[10], row[11], row[12], row[13], row[14])
        cw.close()
        jw = JobLogWriter(os.path.join(self.out_dir, "job_log.csv"))
        n_jl = int(t["jl_count"].item())
        jrows = t["jl_rows"][:n_jl].cpu().numpy()
        from ..models.coeffs import LatencyCoeffs, PowerCoeffs
        from ..policies.gridsearch import energy_tuple
        for row in jrows:
            jid, ing, jt, size, d, fused, n, netlat, start, finish = row
            d = int(d)
            jt = int(jt)
            pC = PowerCoeffs(*self.sc.power_coeffs[d, jt, :])
            tC = LatencyCoeffs(*self.sc.latency_coeffs[d, jt, :])
            T_pred, P_pred, E_pred = energy_tuple(int(n), float(fused), pC, tC)
            jw.row(int(jid), self.sc.ingress_names[int(ing)],
                   "inference" if jt == 0 else "training", float(size),
                   self.sc.dc_names[d], float(fused), int(n), float(netlat),
                   float(start), float(finish), 0, T_pred, P_pred, E_pred)
        jw.close()
