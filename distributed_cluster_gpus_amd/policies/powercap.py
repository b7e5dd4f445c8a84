"""Power-cap "atoms": discrete DVFS steps priced by power-per-throughput.

Behavioral spec (reference simcore/freq_load_agg.py:8-80, re-derived here in
the repo's array-oriented idiom — the GPU engine carries the same math as a
frozen-ladder selection walk in ops/csrc/hip/replica_engine.hip):

* For a running task at frequency f on a sorted ladder of levels, evaluate
  throughput V(level) = 1/T(n, level) and power P(level) at EVERY level once.
* Down-atoms are the consecutive level steps from the task's nearest ladder
  index downward; up-atoms the steps upward.  Each atom's (dV, dP) is the
  clamped consecutive difference of the V/P curves — independent of which
  atoms survive filtering — and rho = dP/dV prices it.  Atoms with zero
  throughput change are dropped.
* ``aggregate_atoms`` merges atoms across tasks sorted ascending by rho,
  STABLY: equal-rho atoms keep (task order, step order) — cap_greedy's
  trajectory depends on this tie-break.
"""
from dataclasses import dataclass
from typing import Iterable, List, Sequence, Tuple

import numpy as np

from ..models.coeffs import LatencyCoeffs, PowerCoeffs
from ..models.latency import unit_time_s
from ..models.power import job_power_w


@dataclass
class RunningTask:
    job_id: int
    dc_name: str
    n: int
    f: float
    freq_levels: List[float]
    pc: PowerCoeffs
    tc: LatencyCoeffs


@dataclass(frozen=True)
class DVFSAtom:
    rho: float      # dP / dV — watts paid (saved) per unit/s gained (lost)
    dV: float
    dP: float
    job_id: int
    dc_name: str
    f_from: float
    f_to: float


def _ladder_curves(t: RunningTask) -> Tuple[Sequence[float], np.ndarray, np.ndarray]:
    """(sorted levels, V[level], P[level]) for one task — each level priced
    once; the scalar model functions keep bit-parity with the engines."""
    lv = sorted(t.freq_levels)
    V = np.array([(lambda T: 0.0 if T <= 0 else 1.0 / T)(unit_time_s(t.n, f, t.tc))
                  for f in lv])
    P = np.array([job_power_w(t.n, f, t.pc) for f in lv])
    return lv, V, P


def _steps(lv, V, P, idx_pairs, jid, dc) -> List[DVFSAtom]:
    """Atoms for consecutive (from, to) ladder index pairs; magnitude-clamped
    diffs, zero-dV steps dropped."""
    out = []
    for i_from, i_to in idx_pairs:
        if i_to > i_from:   # up: gain throughput, pay power
            dV = max(0.0, V[i_to] - V[i_from])
            dP = max(0.0, P[i_to] - P[i_from])
        else:               # down: lose throughput, save power
            dV = max(0.0, V[i_from] - V[i_to])
            dP = max(0.0, P[i_from] - P[i_to])
        if dV > 0:
            out.append(DVFSAtom(dP / dV, dV, dP, jid, dc,
                                lv[i_from], lv[i_to]))
    return out


def atoms_for_task(t: RunningTask):
    """(up_atoms, down_atoms) from the task's current frequency."""
    lv, V, P = _ladder_curves(t)
    i0 = int(np.argmin(np.abs(np.asarray(lv) - t.f)))  # nearest level, first wins
    up = _steps(lv, V, P, [(k, k + 1) for k in range(i0, len(lv) - 1)],
                t.job_id, t.dc_name)
    down = _steps(lv, V, P, [(k, k - 1) for k in range(i0, 0, -1)],
                  t.job_id, t.dc_name)
    return up, down


def aggregate_atoms(tasks: Iterable[RunningTask]):
    """Merge atoms across tasks, ascending-rho with a STABLE order (equal-rho
    atoms keep task-major, step-minor order)."""
    ups, downs = [], []
    for t in tasks:
        u, d = atoms_for_task(t)
        ups.extend(u)
        downs.extend(d)

    def _stable_by_rho(atoms):
        if not atoms:
            return atoms
        order = np.argsort(np.array([a.rho for a in atoms]), kind="stable")
        return [atoms[i] for i in order]

    return _stable_by_rho(ups), _stable_by_rho(downs)
