"""UCB1 bandit over discrete DVFS levels, per (dc, jobtype) context.

Semantics parity with the reference ``BanditDVFS`` (simcore/learners.py:5-43):
round-robin initial exploration, UCB1 selection, reward = -cost_per_unit.
The batched engine keeps the same (N, S, t) statistics as per-replica device
arrays and runs the argmax in the arrival kernel.
"""
import math
from collections import defaultdict


class UCB1DVFS:
    def __init__(self, init_explore: int = 1, objective: str = "energy"):
        self.N = defaultdict(int)      # pull counts per arm
        self.S = defaultdict(float)    # summed rewards per arm
        self.t = 0
        self.objective = objective
        self.init_explore = init_explore

    @staticmethod
    def _key(dc_name, job_type, f):
        return (dc_name, job_type, float(f))

    def select(self, dc_name, job_type, freq_levels):
        self.t += 1
        for f in freq_levels:
            if self.N[self._key(dc_name, job_type, f)] < self.init_explore:
                return f
        best_f, best_ucb = None, -1e9
        for f in freq_levels:
            k = self._key(dc_name, job_type, f)
            n = self.N[k]
            mean = self.S[k] / n if n > 0 else 0.0
            ucb = mean + math.sqrt(2.0 * math.log(self.t) / n)
            if ucb > best_ucb:
                best_ucb, best_f = ucb, f
        return best_f

    def update(self, dc_name, job_type, f, cost_per_unit):
        k = self._key(dc_name, job_type, float(f))
        self.N[k] += 1
        self.S[k] += -float(cost_per_unit)
