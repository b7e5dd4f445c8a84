"""DVFS power model (host scalar form).

Semantics match the reference ``gpu_power_w`` / ``task_power_w``
(reference: simcore/energy_paper.py:4-12).  The device-side form is a fused
HIP device function in ops/csrc/hip/sim_models.hpp.
"""
from .coeffs import PowerCoeffs


def gpu_power_w(f_ghz: float, c: PowerCoeffs) -> float:
    """P_gpu(f) = alpha_p f^3 + beta_p f + gamma_p, with f clamped at 0."""
    f = max(0.0, float(f_ghz))
    return c.alpha_p * (f ** 3) + c.beta_p * f + c.gamma_p


def job_power_w(n_gpus: int, f_ghz: float, c: PowerCoeffs) -> float:
    """Whole-job power: n * P_gpu(f); n coerced to a non-negative int."""
    n = max(0, int(n_gpus))
    return n * gpu_power_w(f_ghz, c)
