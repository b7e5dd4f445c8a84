"""Calibrated analytic-model coefficient types.

Capability parity with the reference dataclasses ``TrainPowerCoeffs`` /
``TrainLatencyCoeffs`` (reference: simcore/coeffs.py:4-17).  Stored as plain
float triples here so they can be packed straight into the flat device-side
coefficient tables the HIP engine consumes (see models/scenario.py).
"""
from dataclasses import dataclass


@dataclass(frozen=True)
class PowerCoeffs:
    """Per-GPU DVFS power model  P(f) = alpha_p * f^3 + beta_p * f + gamma_p  [W]."""
    alpha_p: float
    beta_p: float
    gamma_p: float

    def as_tuple(self):
        return (self.alpha_p, self.beta_p, self.gamma_p)


@dataclass(frozen=True)
class LatencyCoeffs:
    """Per-unit service-time model  T(n,f)  [s/unit]:

        n == 1 : alpha_t + beta_t / f
        n  > 1 : (alpha_t + beta_t / f + gamma_t * n) / n
    """
    alpha_t: float
    beta_t: float
    gamma_t: float

    def as_tuple(self):
        return (self.alpha_t, self.beta_t, self.gamma_t)
