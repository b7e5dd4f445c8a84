"""Policy-layer tests: grid search, bandit, power-cap atoms, heuristics."""
import math

import pytest

from distributed_cluster_gpus_amd.models.cluster import DataCenterState, JobState
from distributed_cluster_gpus_amd.models.coeffs import LatencyCoeffs, PowerCoeffs
from distributed_cluster_gpus_amd.models.latency import unit_time_s
from distributed_cluster_gpus_amd.models.power import job_power_w
from distributed_cluster_gpus_amd.models.scenario import PolicyParams
from distributed_cluster_gpus_amd.policies.bandit import UCB1DVFS
from distributed_cluster_gpus_amd.policies.gridsearch import (
    best_energy_freq, best_nf_grid, energy_tuple, freq_for_perf_expand)
from distributed_cluster_gpus_amd.policies.heuristic import heuristic_allocate
from distributed_cluster_gpus_amd.policies.powercap import (
    RunningTask, aggregate_atoms, atoms_for_task)

PC = PowerCoeffs(75.0, 80.0, 110.0)
TC = LatencyCoeffs(0.0045, 0.032, 0.0012)
LEVELS = [0.3, 0.4, 0.5, 0.6, 0.7, 0.8, 0.9, 1.0]


def brute_force_grid(n_max, levels, pc, tc, objective="energy", ci=0.0,
                     price=0.0, deadline=None):
    best = None
    for n in range(1, n_max + 1):
        for f in levels:
            T, P, E = energy_tuple(n, f, pc, tc)
            if deadline is not None and T > deadline:
                continue
            score = {"energy": E, "carbon": E * ci,
                     "cost": E / 3.6e6 * price}[objective]
            if best is None or score < best[0]:
                best = (score, n, f)
    return best


def test_energy_tuple():
    T, P, E = energy_tuple(4, 0.8, PC, TC)
    assert T == pytest.approx(unit_time_s(4, 0.8, TC))
    assert P == pytest.approx(job_power_w(4, 0.8, PC))
    assert E == pytest.approx(T * P)


def test_best_energy_freq_matches_bruteforce():
    for n in (1, 2, 4, 8):
        f = best_energy_freq(n, LEVELS, PC, TC)
        es = {fl: energy_tuple(n, fl, PC, TC)[2] for fl in LEVELS}
        assert es[f] == min(es.values())


def test_best_nf_grid_matches_bruteforce():
    for obj, ci, price in (("energy", 0, 0), ("carbon", 350.0, 0), ("cost", 0, 0.2)):
        n, f, T, P, E = best_nf_grid(8, LEVELS, PC, TC, objective=obj,
                                     carbon_intensity=ci, price_kwh=price)
        _, bn, bf = brute_force_grid(8, LEVELS, PC, TC, obj, ci, price)
        assert (n, f) == (bn, bf)


def test_best_nf_grid_deadline_filter():
    ddl = 0.02
    n, f, T, P, E = best_nf_grid(8, LEVELS, PC, TC, deadline_s=ddl)
    assert T <= ddl
    _, bn, bf = brute_force_grid(8, LEVELS, PC, TC, deadline=ddl)
    assert (n, f) == (bn, bf)


def test_best_nf_grid_deadline_infeasible_fallback():
    # impossible deadline -> n=1 at f_max, per-GPU power fallback
    n, f, T, P, E = best_nf_grid(8, LEVELS, PC, TC, deadline_s=1e-9)
    assert (n, f) == (1, 1.0)
    from distributed_cluster_gpus_amd.models.power import gpu_power_w
    assert P == pytest.approx(gpu_power_w(1.0, PC))


def test_freq_for_perf_expand():
    f1 = freq_for_perf_expand(2, 0.6, 4, TC, LEVELS)
    assert f1 in LEVELS
    # growing n at same f speeds the job up, so preserving T needs f <= f0
    assert f1 <= 0.6 + 1e-9
    # the chosen ladder step should give roughly the original unit time
    T0 = unit_time_s(2, 0.6, TC)
    T1 = unit_time_s(4, f1, TC)
    assert T1 == pytest.approx(T0, rel=0.35)  # ladder is coarse
    # n1 == 1 branch
    f2 = freq_for_perf_expand(1, 0.6, 1, TC, LEVELS)
    assert f2 == pytest.approx(0.6)


def test_ucb1_explores_then_exploits():
    b = UCB1DVFS(init_explore=1)
    # select/update pairs: each unexplored arm is offered exactly once
    sel = []
    for _ in LEVELS:
        f = b.select("dc", "inference", LEVELS)
        sel.append(f)
        b.update("dc", "inference", f, 1.0 if f != 0.5 else 0.01)
    assert sel == LEVELS  # round-robin exploration first
    picks = []
    for _ in range(50):
        f = b.select("dc", "inference", LEVELS)
        picks.append(f)
        b.update("dc", "inference", f, 1.0 if f != 0.5 else 0.01)
    assert picks.count(0.5) > 25


def test_atoms_ladders():
    t = RunningTask(job_id=1, dc_name="dc", n=2, f=0.6, freq_levels=LEVELS,
                    pc=PC, tc=TC)
    up, down = atoms_for_task(t)
    # from 0.6 there are 4 up steps and 3 down steps
    assert len(up) == 4 and len(down) == 3
    for a in down:
        assert a.f_to < a.f_from
        assert a.dP >= 0 and a.dV > 0
        assert a.rho == pytest.approx(a.dP / a.dV)
    # down-atoms from one task get cheaper (lower rho) at lower frequencies?
    # rho ordering across aggregate must be ascending
    _, down_all = aggregate_atoms([t, RunningTask(2, "dc", 4, 1.0, LEVELS, PC, TC)])
    rhos = [a.rho for a in down_all]
    assert rhos == sorted(rhos)


def _mk_dc(total=8, default_freq=1.0):
    return DataCenterState(name="d", gpu_name="g", p_idle=45.0, p_peak=350.0,
                           p_sleep=28.0, pow_alpha=3.0, total_gpus=total,
                           freq_levels=list(LEVELS), default_freq=default_freq)


def _mk_job(jtype="inference"):
    return JobState(jid=1, ingress="i", jtype=jtype, size=1.0, arrival_time=0.0)


def test_heuristic_energy_aware_training_scales_out():
    dc = _mk_dc()
    pol = PolicyParams(name="energy_aware")
    g = heuristic_allocate(dc, _mk_job("training"), pol)
    assert g == 8 and dc.current_freq == pol.dvfs_low


def test_heuristic_inference_goes_high():
    for pname in ("energy_aware", "perf_first"):
        dc = _mk_dc()
        dc.current_freq = 0.5
        g = heuristic_allocate(dc, _mk_job("inference"), PolicyParams(name=pname))
        assert g == 8 and dc.current_freq == 1.0


def test_heuristic_unknown_policy_raises():
    with pytest.raises(ValueError):
        heuristic_allocate(_mk_dc(), _mk_job(), PolicyParams(name="nope"))


def test_baseline_power_model():
    dc = _mk_dc(total=4)
    dc.busy_gpus = 1
    dc.current_freq = 0.5
    expect = 1 * (45.0 + 350.0 * 0.5 ** 3) + 3 * 28.0
    assert dc.baseline_power_w() == pytest.approx(expect)


def test_inference_lut():
    from distributed_cluster_gpus_amd.policies.lut import InferenceLUT
    lut = InferenceLUT({(1.0, 8): 0.1}, {(1.0, 8): 5.0})
    T, E = lut.time_and_energy(n=2, f=1.0, b=8, l=32)
    assert T == pytest.approx(0.1 * 2)   # ceil(32/16)=2 batches
    assert E == pytest.approx(2 * 2 * 5.0)
    with pytest.raises(KeyError):
        lut.time_and_energy(1, 0.5, 8, 32)
