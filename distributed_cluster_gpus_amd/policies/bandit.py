"""UCB1 bandit over discrete DVFS levels, one arm table per (dc, jobtype).

Behavioral spec (reference simcore/learners.py:5-43, re-derived in the repo's
array idiom — the same (N, S, t) statistics the batched engine keeps as
per-replica device arrays):

* one global pull clock t, incremented per selection;
* round-robin initial exploration: the first level (in the caller's ladder
  order) with fewer than ``init_explore`` pulls is returned as-is;
* otherwise UCB1: argmax over mean + sqrt(2 ln t / n), FIRST maximum in
  ladder order (strict-improvement scan semantics);
* rewards are negated costs-per-unit (lower cost = higher reward).
"""
import numpy as np


class _ArmTable:
    """Pull counts / reward sums for one (dc, jobtype) context, indexed by
    frequency value (levels registered on first sight, in sighting order)."""

    __slots__ = ("slot", "N", "S")

    def __init__(self):
        self.slot = {}
        self.N = np.zeros(0, dtype=np.int64)
        self.S = np.zeros(0, dtype=np.float64)

    def index(self, f: float) -> int:
        k = self.slot.get(float(f))
        if k is None:
            k = len(self.slot)
            self.slot[float(f)] = k
            self.N = np.append(self.N, 0)
            self.S = np.append(self.S, 0.0)
        return k


class UCB1DVFS:
    def __init__(self, init_explore: int = 1, objective: str = "energy"):
        self.t = 0
        self.objective = objective
        self.init_explore = int(init_explore)
        self._ctx = {}

    def _table(self, dc_name, job_type) -> _ArmTable:
        return self._ctx.setdefault((dc_name, job_type), _ArmTable())

    def select(self, dc_name, job_type, freq_levels):
        self.t += 1
        tab = self._table(dc_name, job_type)
        idx = np.array([tab.index(f) for f in freq_levels])
        n = tab.N[idx]
        cold = np.flatnonzero(n < self.init_explore)
        if cold.size:
            return freq_levels[int(cold[0])]
        # UCB1 over the whole ladder at once; np.argmax = first maximum,
        # matching the reference's strict-> scan order
        ucb = tab.S[idx] / n + np.sqrt(2.0 * np.log(self.t) / n)
        return freq_levels[int(np.argmax(ucb))]

    def update(self, dc_name, job_type, f, cost_per_unit):
        tab = self._table(dc_name, job_type)
        k = tab.index(f)
        tab.N[k] += 1
        tab.S[k] += -float(cost_per_unit)

    # introspection used by tests/analysis
    @property
    def N(self):
        return {(dc, jt, f): int(tab.N[k]) for (dc, jt), tab in self._ctx.items()
                for f, k in tab.slot.items()}

    @property
    def S(self):
        return {(dc, jt, f): float(tab.S[k]) for (dc, jt), tab in self._ctx.items()
                for f, k in tab.slot.items()}
