"""Arrival processes and job-size distributions.

Capability parity: reference simcore/arrivals.py.  Two deliberate behavioural
notes, both preserved for log parity (SURVEY.md §6 reproduction note /
Appendix A.1):

* Sinusoid arrivals use thinning whose rejection loop re-derives the candidate
  time from the SAME base ``t`` each iteration (it does not accumulate rejected
  gaps), which biases the effective rate toward rate*(1+|amp|).  The faithful
  form is the default; ``accumulate=True`` gives the textbook-correct thinning.
* Inference sizes are Pareto(x_m=1, alpha=1.8); training sizes are
  lognormal(mu=ln 50000, sigma=0.4) floored at 0.1.

RNG is dependency-injected (any object with the ``random.Random`` API) so the
scalar engines can share one CPython-compatible Mersenne stream.
"""
import math
from dataclasses import dataclass

# job-size distribution constants (reference: simcore/arrivals.py:5-11)
PARETO_XM = 1.0
PARETO_ALPHA = 1.8
LOGNORM_MU = math.log(50000.0)
LOGNORM_SIGMA = 0.4
TRAIN_SIZE_FLOOR = 0.1


def sample_job_size(jtype: str, rng) -> float:
    """Draw a job size in abstract work units for 'inference' or 'training'."""
    if jtype == "inference":
        u = max(1e-9, 1.0 - rng.random())
        return PARETO_XM / (u ** (1.0 / PARETO_ALPHA))
    return max(TRAIN_SIZE_FLOOR, rng.lognormvariate(LOGNORM_MU, LOGNORM_SIGMA))


def _expovariate_safe(rng, lmbda: float) -> float:
    return float("inf") if lmbda <= 0 else rng.expovariate(lmbda)


@dataclass
class ArrivalProcess:
    """Inter-arrival generator: 'poisson' | 'sinusoid' | 'off'."""
    mode: str
    rate: float
    amp: float = 0.0
    period: float = 3600.0
    accumulate: bool = False  # True = textbook thinning (documented divergence)

    def lambda_t(self, t: float) -> float:
        if self.mode == "poisson":
            return self.rate
        if self.mode == "sinusoid":
            return max(0.0, self.rate * (1.0 + self.amp *
                                         math.sin(2.0 * math.pi * (t % self.period) / self.period)))
        if self.mode == "off":
            return 0.0
        raise ValueError(f"Unknown arrival mode {self.mode!r}")

    def next_interarrival(self, t: float, rng) -> float:
        if self.mode == "poisson":
            return _expovariate_safe(rng, self.rate)
        if self.mode == "sinusoid":
            max_rate = self.rate * (1.0 + abs(self.amp))
            elapsed = 0.0
            while True:
                w = _expovariate_safe(rng, max_rate)
                if self.accumulate:
                    elapsed += w
                    if rng.random() <= self.lambda_t(t + elapsed) / max_rate:
                        return elapsed
                else:
                    # faithful non-accumulating rejection (reference arrivals.py:39-44)
                    if rng.random() <= self.lambda_t(t + w) / max_rate:
                        return w
        if self.mode == "off":
            return float("inf")
        raise ValueError(f"Unknown arrival mode {self.mode!r}")
