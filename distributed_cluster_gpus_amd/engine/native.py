"""NativeEngine: Python wrapper around the C++ scalar DES core (_des_core).

Bitwise log parity with the oracle (tests/test_native_engine.py asserts
byte-identical CSVs) at 1-2 orders of magnitude higher events/sec.  The 8
non-RL algorithms run fully native; chsac_af needs torch in the loop and
transparently falls back to the OracleEngine (documented — the batched MI355X
engine is the scale path for RL).
"""
import os
from typing import Optional

from ..models.arrivals import ArrivalProcess
from ..models.scenario import PAYLOAD_GB, Scenario
from ..ops import load_des_core
from .oracle import ALGOS, OracleEngine

_ALGO_IDS = {"default_policy": 0, "cap_uniform": 1, "cap_greedy": 2,
             "joint_nf": 3, "bandit": 4, "carbon_cost": 5, "eco_route": 6,
             "debug": 7}
_ECO_IDS = {"energy": 0, "carbon": 1, "cost": 2}


class NativeEngine:
    def __init__(self, scenario: Scenario,
                 arrival_inf: ArrivalProcess, arrival_trn: ArrivalProcess,
                 *, algo: str = "default_policy",
                 duration: float = 3600.0, log_interval: float = 10.0,
                 out_dir: Optional[str] = None, seed: int = 42,
                 power_cap: float = 0.0, control_interval: float = 5.0,
                 elastic_scaling: bool = False, eco_objective: str = "energy",
                 num_fixed_gpus: int = 1, fixed_freq: Optional[float] = None,
                 use_control_interval: bool = False,
                 logger=None, show_progress: bool = False, **rl_kwargs):
        if algo not in ALGOS:
            raise ValueError(f"unknown algo {algo!r}")
        self.rl = None
        self._fallback = None
        if algo == "chsac_af":
            # RL-in-the-loop stays on the torch path
            self._fallback = OracleEngine(
                scenario, arrival_inf, arrival_trn, algo=algo,
                duration=duration, log_interval=log_interval, out_dir=out_dir,
                seed=seed, power_cap=power_cap, control_interval=control_interval,
                use_control_interval=use_control_interval,
                elastic_scaling=elastic_scaling, eco_objective=eco_objective,
                num_fixed_gpus=num_fixed_gpus, fixed_freq=fixed_freq,
                logger=logger, show_progress=show_progress, **rl_kwargs)
            self.rl = self._fallback.rl
            return

        core = load_des_core()
        sc = scenario
        sc_dict = {
            "n_dc": sc.n_dc, "n_ing": sc.n_ing,
            "dc_names": list(sc.dc_names),
            "total_gpus": [int(x) for x in sc.total_gpus],
            "p_idle": [float(x) for x in sc.p_idle],
            "p_sleep": [float(x) for x in sc.p_sleep],
            "power_gating": [int(x) for x in sc.power_gating],
            "freq_levels": [float(x) for x in sc.freq_levels],
            "default_freq": [float(x) for x in sc.default_freq],
            "power_coeffs": [float(x) for x in sc.power_coeffs.reshape(-1)],
            "latency_coeffs": [float(x) for x in sc.latency_coeffs.reshape(-1)],
            "wan_latency_s": [float(x) for row in sc.wan_latency_s for x in row],
            "wan_bottleneck_gbps": [float(x) for row in sc.wan_bottleneck_gbps for x in row],
            "carbon": [float(x) for x in sc.carbon_vec()],
            "price24": [float(x) for x in sc.price_vec24()],
            "payload_inf_gb": PAYLOAD_GB[0], "payload_trn_gb": PAYLOAD_GB[1],
            "policy_name": sc.policy.name,
            "max_gpus_per_job": int(sc.policy.max_gpus_per_job),
            "inf_priority": bool(sc.policy.inf_priority),
            "dvfs_low": float(sc.policy.dvfs_low),
            "dvfs_high": float(sc.policy.dvfs_high),
            "train_scale_out_low_freq": bool(sc.policy.train_scale_out_low_freq),
        }
        out_dir = out_dir or os.getcwd()
        os.makedirs(out_dir, exist_ok=True)
        params = {
            "algo": _ALGO_IDS[algo],
            "duration": float(duration), "log_interval": float(log_interval),
            "seed": int(seed), "power_cap": float(power_cap),
            "eco_objective": _ECO_IDS[eco_objective],
            "num_fixed_gpus": int(num_fixed_gpus),
            "fixed_freq": float(fixed_freq) if fixed_freq else 0.0,
            "cluster_csv": os.path.join(out_dir, "cluster_log.csv"),
            "job_csv": os.path.join(out_dir, "job_log.csv"),
            "arr_inf": {"mode": arrival_inf.mode, "rate": float(arrival_inf.rate),
                        "amp": float(arrival_inf.amp), "period": float(arrival_inf.period)},
            "arr_trn": {"mode": arrival_trn.mode, "rate": float(arrival_trn.rate),
                        "amp": float(arrival_trn.amp), "period": float(arrival_trn.period)},
        }
        self._sim = core.DesSim(sc_dict, params)
        self._sim.set_ingress_names(list(sc.ingress_names))
        self._sim.set_baseline([sc.gpu_specs[n].p_peak for n in sc.dc_names],
                               [sc.gpu_specs[n].alpha for n in sc.dc_names])

    def run(self):
        if self._fallback is not None:
            return self._fallback.run()
        return dict(self._sim.run())
