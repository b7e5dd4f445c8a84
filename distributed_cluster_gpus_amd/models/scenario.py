"""Scenario: a fully-resolved, numeric description of one simulation setup.

This is the single hand-off structure between the config layer and every
engine.  The scalar engines read the object fields; the batched MI355X engine
packs the numpy tables directly into device buffers (coefficients become the
constant tables the HIP step kernels read — SURVEY §2 rows 6/9).

Everything is indexed, not named:  dc index d in [0, n_dc), ingress index i in
[0, n_ing), job type j in {0=inference, 1=training}, frequency level k in
[0, n_freq).
"""
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from .arrivals import ArrivalProcess
from .cluster import DataCenterState
from .gputypes import GPUSpec
from .wan import IngressSpec, WanGraph, dijkstra_tables

# WAN payload sizes by job type (GB); reference hardcodes these at
# simulator_paper_multi.py:489.
PAYLOAD_GB = (0.05, 5.0)  # (inference, training)


@dataclass
class PolicyParams:
    """Heuristic in-DC allocator knobs (reference PolicyConfig, simcore/policy.py:5-13)."""
    name: str = "energy_aware"           # 'energy_aware' | 'perf_first'
    max_gpus_per_job: int = 8
    inf_priority: bool = True
    dvfs_low: float = 0.6
    dvfs_high: float = 1.0
    train_scale_out_low_freq: bool = True
    reserve_inf_gpus: int = 0


@dataclass
class RouterParams:
    """DC-choice score weights (reference RouterPolicy, simcore/router.py:3-9).
    The reference stores these but routes uniformly at random for non-eco/RL
    algos; we keep that default and expose the weighted score as an opt-in
    (`use_weighted=True`) capability extension."""
    w_energy: float = 1.0
    w_latency: float = 0.5
    w_carbon: float = 0.0
    d_choices: int = 0
    use_weighted: bool = False


@dataclass
class Scenario:
    dc_names: List[str]
    ingress_names: List[str]
    gpu_specs: Dict[str, GPUSpec]                  # dc_name -> spec
    total_gpus: np.ndarray                         # [n_dc] int32
    freq_levels: np.ndarray                        # [n_freq] float64 (shared ladder)
    default_freq: np.ndarray                       # [n_dc] float64
    power_gating: np.ndarray                       # [n_dc] bool
    # analytic model coefficient tables, indexed [n_dc, 2(jtype), 3]
    power_coeffs: np.ndarray                       # alpha_p, beta_p, gamma_p
    latency_coeffs: np.ndarray                     # alpha_t, beta_t, gamma_t
    # idle-path power per DC: [n_dc] p_idle / p_sleep
    p_idle: np.ndarray
    p_sleep: np.ndarray
    # WAN tables, indexed [n_ing, n_dc]
    wan_latency_s: np.ndarray
    wan_bottleneck_gbps: np.ndarray
    wan_cost_per_gb: np.ndarray
    wan_paths: dict
    graph: WanGraph
    ingresses: Dict[str, IngressSpec]
    # economics
    carbon_intensity: Dict[str, float]             # dc_name -> gCO2/kWh (sparse)
    energy_price_hourly: Dict[int, float]          # hour -> USD/kWh
    # knobs
    policy: PolicyParams = field(default_factory=PolicyParams)
    router: RouterParams = field(default_factory=RouterParams)
    arrival_inf: Optional[ArrivalProcess] = None
    arrival_trn: Optional[ArrivalProcess] = None

    @property
    def n_dc(self) -> int:
        return len(self.dc_names)

    @property
    def n_ing(self) -> int:
        return len(self.ingress_names)

    @property
    def n_freq(self) -> int:
        return int(self.freq_levels.shape[0])

    def carbon_vec(self) -> np.ndarray:
        return np.array([self.carbon_intensity.get(n, 0.0) for n in self.dc_names])

    def price_vec24(self) -> np.ndarray:
        return np.array([float(self.energy_price_hourly.get(h, 0.0)) for h in range(24)])

    def make_dc_states(self) -> Dict[str, DataCenterState]:
        """Materialize fresh mutable DataCenterState objects for a scalar run."""
        out: Dict[str, DataCenterState] = {}
        for d, name in enumerate(self.dc_names):
            spec = self.gpu_specs[name]
            out[name] = DataCenterState(
                name=name, gpu_name=spec.name,
                p_idle=spec.p_idle, p_peak=spec.p_peak, p_sleep=spec.p_sleep,
                pow_alpha=spec.alpha,
                total_gpus=int(self.total_gpus[d]),
                freq_levels=[float(f) for f in self.freq_levels],
                default_freq=float(self.default_freq[d]),
                power_gating=bool(self.power_gating[d]),
            )
        return out


def build_scenario(dc_names, ingress_names, gpu_specs, total_gpus, freq_levels,
                   default_freq, power_gating, coeffs_map, graph, ingresses,
                   carbon, price_hourly, policy=None, router=None) -> Scenario:
    """Assemble a Scenario from name-keyed config pieces.

    coeffs_map: {(dc_name, 'inference'|'training'): (PowerCoeffs, LatencyCoeffs)}
    """
    n_dc = len(dc_names)
    pc = np.zeros((n_dc, 2, 3))
    lc = np.zeros((n_dc, 2, 3))
    for d, name in enumerate(dc_names):
        for j, jname in enumerate(("inference", "training")):
            p, t = coeffs_map[(name, jname)]
            pc[d, j, :] = p.as_tuple()
            lc[d, j, :] = t.as_tuple()
    lat, bw, cost, paths = dijkstra_tables(graph, list(ingress_names), list(dc_names))
    return Scenario(
        dc_names=list(dc_names), ingress_names=list(ingress_names),
        gpu_specs=dict(gpu_specs),
        total_gpus=np.asarray(total_gpus, dtype=np.int32),
        freq_levels=np.asarray(freq_levels, dtype=np.float64),
        default_freq=np.asarray(default_freq, dtype=np.float64),
        power_gating=np.asarray(power_gating, dtype=bool),
        power_coeffs=pc, latency_coeffs=lc,
        p_idle=np.array([gpu_specs[n].p_idle for n in dc_names]),
        p_sleep=np.array([gpu_specs[n].p_sleep for n in dc_names]),
        wan_latency_s=np.asarray(lat), wan_bottleneck_gbps=np.asarray(bw),
        wan_cost_per_gb=np.asarray(cost), wan_paths=paths,
        graph=graph, ingresses=dict(ingresses),
        carbon_intensity=dict(carbon), energy_price_hourly=dict(price_hourly),
        policy=policy or PolicyParams(), router=router or RouterParams(),
    )
