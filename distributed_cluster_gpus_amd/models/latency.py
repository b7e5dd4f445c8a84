"""Per-unit service-time model (host scalar form).

Semantics match the reference ``step_time_s`` (reference:
simcore/latency_paper.py:4-9).  Device-side form lives in
ops/csrc/hip/sim_models.hpp (fused into the step kernels).
"""
from .coeffs import LatencyCoeffs


def unit_time_s(n_gpus: int, f_ghz: float, c: LatencyCoeffs) -> float:
    """Seconds per work unit at n GPUs and frequency f (f floored at 1e-9)."""
    n = max(1, int(n_gpus))
    f = max(1e-9, float(f_ghz))
    if n == 1:
        return c.alpha_t + c.beta_t / f
    return (c.alpha_t + c.beta_t / f + c.gamma_t * n) / n
