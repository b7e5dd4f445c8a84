"""The canonical multi-DC "paper" scenario.

Every numeric value here reproduces the reference configuration exactly
(reference: configs/paper_config.py:20-300) — 8 DCs / 880 GPUs, 8 frequency
levels 0.3..1.0, 8 ingress gateways with an asymmetric WAN, per-(DC, jobtype)
cubic power and latency coefficients, carbon intensity for 3 DCs only, and a
global hourly tariff.  Log parity with the reference depends on these values,
so do not "fix" anything here (including the WAN edge that the reference adds
twice — see _WAN_EDGES below and SURVEY Appendix A.10).

The data is laid out as flat tables (rather than repeated constructor calls)
because the batched MI355X engine consumes it as packed arrays via
models.scenario.build_scenario.
"""
from ..models.arrivals import ArrivalProcess
from ..models.coeffs import LatencyCoeffs, PowerCoeffs
from ..models.gputypes import GPUSpec
from ..models.scenario import PolicyParams, RouterParams, Scenario, build_scenario
from ..models.wan import IngressSpec, WanGraph

FREQ_LADDER = [0.3, 0.4, 0.5, 0.6, 0.7, 0.8, 0.9, 1.0]

# GPU power specs (reference paper_config.py:21-36)
_GPU_SPECS = {
    "A100-SXM4": GPUSpec("A100-SXM4", 50.0, 400.0, 30.0),
    "A100-PCIe": GPUSpec("A100-PCIe", 45.0, 300.0, 28.0),
    "H100-SXM5": GPUSpec("H100-SXM5", 55.0, 700.0, 35.0),
    "H100-PCIe": GPUSpec("H100-PCIe", 45.0, 350.0, 28.0),
    "H200-SXM":  GPUSpec("H200-SXM", 60.0, 700.0, 38.0),
    "H200-PCIe": GPUSpec("H200-PCIe", 55.0, 600.0, 35.0),
    "L4":        GPUSpec("L4", 15.0, 72.0, 8.0),
    "T4":        GPUSpec("T4", 10.0, 70.0, 6.0),
    "A10":       GPUSpec("A10", 20.0, 150.0, 10.0),
    "A30":       GPUSpec("A30", 25.0, 165.0, 12.0),
    "A40":       GPUSpec("A40", 40.0, 300.0, 25.0),
    "L40":       GPUSpec("L40", 35.0, 300.0, 20.0),
    "L40S":      GPUSpec("L40S", 40.0, 350.0, 25.0),
}

# dc_name -> (gpu type, count)    (reference paper_config.py:39-64)
_DC_TABLE = [
    ("us-west",      "H100-PCIe", 16),
    ("us-east",      "A100-PCIe", 32),
    ("eu-west",      "L40S",      256),
    ("eu-central",   "H100-SXM5", 16),
    ("ap-southeast", "L4",        128),
    ("ap-northeast", "H200-PCIe", 16),
    ("sa-east",      "A30",       512),
    ("me-central",   "A10",       512),
]

# (dc, jtype) -> ((alpha_p, beta_p, gamma_p), (alpha_t, beta_t, gamma_t))
# (reference paper_config.py:81-168)
_COEFFS = {
    ("us-west", "training"):       ((75.0, 80.0, 110.0), (0.0045, 0.032, 0.0012)),
    ("us-west", "inference"):      ((95.0, 20.0, 97.0),  (0.0090, 0.0018, 0.0007)),
    ("us-east", "training"):       ((65.0, 60.0, 90.0),  (0.0050, 0.038, 0.0014)),
    ("us-east", "inference"):      ((85.0, 18.0, 80.0),  (0.0080, 0.0020, 0.0009)),
    ("eu-west", "training"):       ((55.0, 40.0, 70.0),  (0.0060, 0.045, 0.0018)),
    ("eu-west", "inference"):      ((70.0, 15.0, 60.0),  (0.0050, 0.020, 0.0010)),
    ("eu-central", "training"):    ((90.0, 85.0, 120.0), (0.0042, 0.030, 0.0011)),
    ("eu-central", "inference"):   ((100.0, 22.0, 100.0), (0.0085, 0.0017, 0.0007)),
    ("ap-southeast", "training"):  ((45.0, 20.0, 40.0),  (0.0065, 0.060, 0.0022)),
    ("ap-southeast", "inference"): ((40.0, 12.0, 35.0),  (0.0045, 0.025, 0.0012)),
    ("ap-northeast", "training"):  ((95.0, 90.0, 125.0), (0.0040, 0.029, 0.0010)),
    ("ap-northeast", "inference"): ((105.0, 25.0, 105.0), (0.0080, 0.0016, 0.0006)),
    ("sa-east", "training"):       ((50.0, 35.0, 65.0),  (0.0062, 0.050, 0.0019)),
    ("sa-east", "inference"):      ((65.0, 14.0, 55.0),  (0.0055, 0.022, 0.0011)),
    ("me-central", "training"):    ((40.0, 25.0, 50.0),  (0.0068, 0.055, 0.0023)),
    ("me-central", "inference"):   ((55.0, 12.0, 45.0),  (0.0050, 0.023, 0.0012)),
}

# ingress gateways (reference paper_config.py:184-193)
_INGRESSES = [
    ("gw-us-west", "US"), ("gw-us-east", "US"),
    ("gw-eu-west", "EU"), ("gw-eu-central", "EU"),
    ("gw-ap-southeast", "APAC"), ("gw-ap-northeast", "APAC"),
    ("gw-sa-east", "SA"), ("gw-me-central", "ME"),
]

# Directed WAN edges (u, v, latency_ms), exactly the reference's insertion order
# (paper_config.py:198-277).  NOTE: the reference adds gw-us-west -> eu-central
# twice (its lines 205-206); we keep the duplicate for graph-structure parity —
# Dijkstra results are unchanged by it.
_WAN_EDGES = [
    ("gw-us-west", "us-west", 12), ("us-west", "gw-us-west", 12),
    ("gw-us-west", "us-east", 70), ("us-east", "gw-us-west", 70),
    ("gw-us-west", "eu-central", 110), ("gw-us-west", "eu-central", 110),
    ("eu-central", "gw-us-west", 110),
    ("gw-us-west", "ap-southeast", 150), ("ap-southeast", "gw-us-west", 150),

    ("gw-us-east", "us-east", 10), ("us-east", "gw-us-east", 10),
    ("gw-us-east", "us-west", 70), ("us-west", "gw-us-east", 70),
    ("gw-us-east", "eu-west", 90), ("eu-west", "gw-us-east", 90),
    ("gw-us-east", "sa-east", 110), ("sa-east", "gw-us-east", 110),

    ("gw-eu-west", "eu-west", 10), ("eu-west", "gw-eu-west", 10),
    ("gw-eu-west", "eu-central", 20), ("eu-central", "gw-eu-west", 20),
    ("gw-eu-west", "us-east", 90), ("us-east", "gw-eu-west", 90),
    ("gw-eu-west", "ap-northeast", 190), ("ap-northeast", "gw-eu-west", 190),

    ("gw-eu-central", "eu-central", 10), ("eu-central", "gw-eu-central", 10),
    ("gw-eu-central", "me-central", 60), ("me-central", "gw-eu-central", 60),
    ("gw-eu-central", "ap-southeast", 170), ("ap-southeast", "gw-eu-central", 170),

    ("gw-ap-southeast", "ap-southeast", 8), ("ap-southeast", "gw-ap-southeast", 8),
    ("gw-ap-southeast", "ap-northeast", 60), ("ap-northeast", "gw-ap-southeast", 60),
    ("gw-ap-southeast", "eu-central", 170), ("eu-central", "gw-ap-southeast", 170),

    ("gw-ap-northeast", "ap-northeast", 8), ("ap-northeast", "gw-ap-northeast", 8),
    ("gw-ap-northeast", "us-west", 130), ("us-west", "gw-ap-northeast", 130),
    ("gw-ap-northeast", "eu-west", 190), ("eu-west", "gw-ap-northeast", 190),

    ("gw-sa-east", "sa-east", 12), ("sa-east", "gw-sa-east", 12),
    ("gw-sa-east", "us-east", 110), ("us-east", "gw-sa-east", 110),
    ("gw-sa-east", "eu-west", 150), ("eu-west", "gw-sa-east", 150),

    ("gw-me-central", "me-central", 10), ("me-central", "gw-me-central", 10),
    ("gw-me-central", "eu-central", 60), ("eu-central", "gw-me-central", 60),
    ("gw-me-central", "ap-southeast", 120), ("ap-southeast", "gw-me-central", 120),
]

# Carbon intensity, gCO2/kWh — deliberately only 3 DCs, like the reference
# (paper_config.py:280-286); other DCs read 0.0.
_CARBON = {"us-west": 350.0, "eu-central": 220.0, "ap-southeast": 500.0}


def _price_hourly():
    """Global USD/kWh tariff by hour (reference paper_config.py:294-300)."""
    p = {}
    for h in range(24):
        p[h] = 0.12 if h < 7 else (0.20 if h < 19 else 0.16)
    return p


# Display-name maps consumed by the per-DC analysis plots
# (reference paper_config.py:303-323).
DC_GPUS_LABEL = {
    "us-west": "16 x H100-PCIe", "us-east": "32 x A100-PCIe",
    "eu-west": "256 x L40S", "eu-central": "16 x H100-SXM",
    "ap-southeast": "128 x L4", "ap-northeast": "16 x H200-PCIe",
    "sa-east": "512 x A30", "me-central": "512 x A10",
}
GW_ALPHABET_LABEL = {
    "gw-us-west": "A", "gw-us-east": "B", "gw-eu-west": "E", "gw-eu-central": "F",
    "gw-ap-southeast": "G", "gw-ap-northeast": "H", "gw-sa-east": "C", "gw-me-central": "D",
}


def _coeffs_map():
    return {k: (PowerCoeffs(*pv), LatencyCoeffs(*tv)) for k, (pv, tv) in _COEFFS.items()}


def paper_scenario(policy: PolicyParams = None, router: RouterParams = None) -> Scenario:
    """The full 8-DC / 8-ingress paper scenario."""
    dc_names = [name for name, _, _ in _DC_TABLE]
    gpu_specs = {name: _GPU_SPECS[gt] for name, gt, _ in _DC_TABLE}
    total = [n for _, _, n in _DC_TABLE]
    graph = WanGraph()
    for u, v, ms in _WAN_EDGES:
        graph.add_edge(u, v, ms)
    ingresses = {n: IngressSpec(n, r) for n, r in _INGRESSES}
    return build_scenario(
        dc_names=dc_names, ingress_names=[n for n, _ in _INGRESSES],
        gpu_specs=gpu_specs, total_gpus=total,
        freq_levels=FREQ_LADDER, default_freq=[1.0] * len(dc_names),
        power_gating=[True] * len(dc_names),
        coeffs_map=_coeffs_map(), graph=graph, ingresses=ingresses,
        carbon=_CARBON, price_hourly=_price_hourly(),
        policy=policy or PolicyParams(),
        router=router or RouterParams(w_energy=1.0, w_latency=0.5, w_carbon=0.0, d_choices=0),
    )


def single_dc_scenario(policy: PolicyParams = None, router: RouterParams = None) -> Scenario:
    """Single-DC debug variant: 128x H100-PCIe behind one gateway
    (reference paper_config.py:10-17, 171-180; used with the debug algo)."""
    graph = WanGraph()
    graph.add_edge("gw-us-west", "us-west", 12)
    graph.add_edge("us-west", "gw-us-west", 12)
    ingresses = {"gw-us-west": IngressSpec("gw-us-west", "US")}
    cmap = _coeffs_map()
    coeffs = {("us-west", j): cmap[("us-west", j)] for j in ("inference", "training")}
    return build_scenario(
        dc_names=["us-west"], ingress_names=["gw-us-west"],
        gpu_specs={"us-west": _GPU_SPECS["H100-PCIe"]}, total_gpus=[128],
        freq_levels=FREQ_LADDER, default_freq=[1.0], power_gating=[True],
        coeffs_map=coeffs, graph=graph, ingresses=ingresses,
        carbon={}, price_hourly=_price_hourly(),
        policy=policy or PolicyParams(),
        router=router or RouterParams(),
    )


def build_arrivals(inf_mode="sinusoid", inf_rate=6.0, inf_amp=0.6, inf_period=300.0,
                   trn_mode="poisson", trn_rate=0.3):
    """Default arrival pair (reference paper_config.py:67-71)."""
    return (ArrivalProcess(mode=inf_mode, rate=inf_rate, amp=inf_amp, period=inf_period),
            ArrivalProcess(mode=trn_mode, rate=trn_rate))
