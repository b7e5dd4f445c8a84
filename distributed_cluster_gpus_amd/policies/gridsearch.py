"""Grid-search (n, f) optimizers over the DVFS ladder.

Semantics parity with the reference ``best_energy_freq`` / ``best_nf_grid`` /
``energy_tuple`` (simcore/policy_paper.py:7-77).  On the batched MI355X path
the same 64-candidate scan is a wavefront argmin inside the step kernel
(ops/csrc/hip/): one lane per (n, f) candidate, 64 lanes = the full 8x8 grid —
a natural fit for CDNA4's 64-wide wavefront.
"""
from typing import Iterable, Optional, Tuple

from ..models.coeffs import LatencyCoeffs, PowerCoeffs
from ..models.latency import unit_time_s
from ..models.power import gpu_power_w, job_power_w

J_PER_KWH = 3.6e6


def energy_tuple(n: int, f: float, pc: PowerCoeffs, tc: LatencyCoeffs) -> Tuple[float, float, float]:
    """(T seconds/unit, P watts, E joules/unit) at (n, f)."""
    T = unit_time_s(n, f, tc)
    P = job_power_w(n, f, pc)
    return (T, P, P * T)


def best_energy_freq(n: int, freq_levels: Iterable[float],
                     pc: PowerCoeffs, tc: LatencyCoeffs) -> float:
    """argmin_f E(n, f); first minimum wins on ties (scan order = given order)."""
    best_f, best_e = None, float("inf")
    for f in freq_levels:
        _, _, E = energy_tuple(n, f, pc, tc)
        if E < best_e:
            best_e, best_f = E, f
    return best_f if best_f is not None else max(freq_levels)


def best_nf_grid(n_max: int, freq_levels, pc: PowerCoeffs, tc: LatencyCoeffs,
                 objective: str = "energy", carbon_intensity: float = 0.0,
                 price_kwh: float = 0.0, deadline_s: Optional[float] = None):
    """argmin over the n x f grid of an energy/carbon/cost score, with an
    optional per-unit deadline filter.  Returns (n*, f*, T*, P*, E*).

    Scan order is n-major then the given frequency order, first minimum wins —
    this tie-break order is part of log parity with the reference.
    """
    best = None
    for n in range(1, max(1, int(n_max)) + 1):
        for f in freq_levels:
            T, P, E = energy_tuple(n, f, pc, tc)
            if deadline_s is not None and T > deadline_s:
                continue
            if objective == "carbon":
                score = E * carbon_intensity
            elif objective == "cost":
                score = (E / J_PER_KWH) * float(price_kwh)
            else:  # "energy" and any unknown objective
                score = E
            if best is None or score < best[0]:
                best = (score, n, f, T, P, E)
    if best is None:
        # nothing met the deadline: fall back to n=1 at max frequency; note the
        # fallback power is per-GPU (not n*P) — reference behaviour
        # (policy_paper.py:70-74) kept for parity.
        fmax = max(freq_levels)
        T = unit_time_s(1, fmax, tc)
        P = gpu_power_w(fmax, pc)
        return 1, fmax, T, P, P * T
    _, n, f, T, P, E = best
    return n, f, T, P, E


def freq_for_perf_expand(n0: int, f0: float, n1: int, tc: LatencyCoeffs,
                         freq_levels: Iterable[float]) -> float:
    """Pick the ladder frequency that preserves T(n0, f0) after growing to n1
    GPUs.  Capability parity with the reference keep_perf_when_expand
    (policy_paper.py:19-29 — dead code there), but with the T(n, f) inversion
    done correctly for n1 > 1: beta/f = T*n - alpha - gamma*n (the reference's
    formula omits the *n on T, valid only at n1 == 1)."""
    n1 = max(1, int(n1))
    T_target = unit_time_s(n0, max(1e-9, f0), tc)
    if n1 == 1:
        denom = T_target - tc.alpha_t
    else:
        denom = T_target * n1 - tc.alpha_t - tc.gamma_t * n1
    if denom <= 1e-12:
        return f0
    f_cont = tc.beta_t / denom
    levels = list(freq_levels)
    return min(levels, key=lambda x: abs(x - f_cont))
