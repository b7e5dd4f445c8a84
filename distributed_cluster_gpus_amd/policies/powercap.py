"""Power-cap "atoms": discrete DVFS steps ranked by power-per-throughput.

Semantics parity with the reference TaskState/Atom/atoms_for_task/
aggregate_with_atoms (simcore/freq_load_agg.py:8-80).  An atom is one discrete
frequency step for one running task; rho = dP/dV prices it (V = units/s
throughput).  The cap_greedy controller applies down-atoms cheapest-rho-first.
"""
from dataclasses import dataclass
from typing import Iterable, List

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


@dataclass
class DVFSAtom:
    rho: float      # dP / dV
    dV: float
    dP: float
    job_id: int
    dc_name: str
    f_from: float
    f_to: float


def _throughput(n, f, tc):
    T = unit_time_s(n, f, tc)
    return 0.0 if T <= 0 else 1.0 / T


def _nearest_idx(levels, f):
    return min(range(len(levels)), key=lambda i: abs(levels[i] - f))


def atoms_for_task(t: RunningTask):
    """Build the up-ladder and down-ladder of atoms from the task's current f."""
    lv = sorted(t.freq_levels)
    i0 = _nearest_idx(lv, t.f)
    v0 = _throughput(t.n, lv[i0], t.tc)
    p0 = job_power_w(t.n, lv[i0], t.pc)
    up, down = [], []

    cur_v, cur_p = v0, p0
    for k in range(i0, len(lv) - 1):
        f_from, f_to = lv[k], lv[k + 1]
        v2 = _throughput(t.n, f_to, t.tc)
        p2 = job_power_w(t.n, f_to, t.pc)
        dV, dP = max(0.0, v2 - cur_v), max(0.0, p2 - cur_p)
        if dV > 0 and dP >= 0:
            up.append(DVFSAtom(dP / dV, dV, dP, t.job_id, t.dc_name, f_from, f_to))
        cur_v, cur_p = v2, p2

    cur_v, cur_p = v0, p0
    for k in range(i0, 0, -1):
        f_from, f_to = lv[k], lv[k - 1]
        v2 = _throughput(t.n, f_to, t.tc)
        p2 = job_power_w(t.n, f_to, t.pc)
        dV, dP = max(0.0, cur_v - v2), max(0.0, cur_p - p2)
        if dV > 0 and dP >= 0:
            down.append(DVFSAtom(dP / dV, dV, dP, t.job_id, t.dc_name, f_from, f_to))
        cur_v, cur_p = v2, p2
    return up, down


def aggregate_atoms(tasks: Iterable[RunningTask]):
    """Merge atoms across tasks, sorted ascending by rho (stable, so equal-rho
    atoms keep task order — matters for cap_greedy log parity)."""
    up_all, down_all = [], []
    for t in tasks:
        u, d = atoms_for_task(t)
        up_all.extend(u)
        down_all.extend(d)
    up_all.sort(key=lambda a: a.rho)
    down_all.sort(key=lambda a: a.rho)
    return up_all, down_all
