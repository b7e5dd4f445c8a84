"""Unit tests for the closed-form analytic models (SURVEY §4 strategy (a))."""
import math
import random

import numpy as np
import pytest

from distributed_cluster_gpus_amd.models.arrivals import (
    ArrivalProcess, LOGNORM_MU, LOGNORM_SIGMA, PARETO_ALPHA, sample_job_size)
from distributed_cluster_gpus_amd.models.coeffs import LatencyCoeffs, PowerCoeffs
from distributed_cluster_gpus_amd.models.latency import unit_time_s
from distributed_cluster_gpus_amd.models.power import gpu_power_w, job_power_w
from distributed_cluster_gpus_amd.models.gputypes import GPUSpec, validate_gpu_specs

PC = PowerCoeffs(75.0, 80.0, 110.0)
TC = LatencyCoeffs(0.0045, 0.032, 0.0012)


def test_power_polynomial():
    for f in (0.0, 0.3, 0.5, 1.0, 1.7):
        assert gpu_power_w(f, PC) == pytest.approx(75 * f ** 3 + 80 * f + 110)
    # negative f clamps to 0
    assert gpu_power_w(-1.0, PC) == pytest.approx(110.0)


def test_job_power_scales_with_n():
    assert job_power_w(4, 0.8, PC) == pytest.approx(4 * gpu_power_w(0.8, PC))
    assert job_power_w(-3, 0.8, PC) == 0.0
    assert job_power_w(2.9, 0.8, PC) == pytest.approx(2 * gpu_power_w(0.8, PC))


def test_latency_single_vs_multi_gpu():
    f = 0.8
    assert unit_time_s(1, f, TC) == pytest.approx(TC.alpha_t + TC.beta_t / f)
    n = 4
    expect = (TC.alpha_t + TC.beta_t / f + TC.gamma_t * n) / n
    assert unit_time_s(n, f, TC) == pytest.approx(expect)
    # f floor
    assert math.isfinite(unit_time_s(1, 0.0, TC))
    # n floor at 1
    assert unit_time_s(0, f, TC) == unit_time_s(1, f, TC)


def test_latency_monotonic_in_f():
    ts = [unit_time_s(1, f, TC) for f in (0.3, 0.5, 0.7, 1.0)]
    assert ts == sorted(ts, reverse=True)


def test_pareto_size_moments():
    rng = random.Random(0)
    xs = [sample_job_size("inference", rng) for _ in range(200000)]
    # Pareto(xm=1, alpha=1.8) mean = alpha/(alpha-1) = 2.25
    assert np.mean(xs) == pytest.approx(PARETO_ALPHA / (PARETO_ALPHA - 1), rel=0.08)
    assert min(xs) >= 1.0


def test_lognormal_size_moments():
    rng = random.Random(0)
    xs = [sample_job_size("training", rng) for _ in range(50000)]
    # lognormal median = exp(mu) = 50000
    assert np.median(xs) == pytest.approx(math.exp(LOGNORM_MU), rel=0.02)
    expected_mean = math.exp(LOGNORM_MU + LOGNORM_SIGMA ** 2 / 2)
    assert np.mean(xs) == pytest.approx(expected_mean, rel=0.02)


def test_poisson_interarrival_rate():
    rng = random.Random(1)
    ap = ArrivalProcess(mode="poisson", rate=4.0)
    xs = [ap.next_interarrival(0.0, rng) for _ in range(100000)]
    assert np.mean(xs) == pytest.approx(1.0 / 4.0, rel=0.02)


def test_sinusoid_thinning_faithful_bias():
    """The faithful (non-accumulating) thinning over-generates: effective rate
    approaches rate*(1+|amp|) (SURVEY §6 reproduction note)."""
    rng = random.Random(2)
    ap = ArrivalProcess(mode="sinusoid", rate=6.0, amp=0.6, period=300.0)
    t, n = 0.0, 0
    while t < 2000.0:
        t += ap.next_interarrival(t, rng)
        n += 1
    eff_rate = n / 2000.0
    assert eff_rate > 6.0 * 1.25          # clearly above nominal
    assert eff_rate < 6.0 * 1.65          # but below the lambda_max ceiling


def test_sinusoid_thinning_accumulate_correct():
    """Textbook thinning (accumulate=True) reproduces the nominal mean rate."""
    rng = random.Random(3)
    ap = ArrivalProcess(mode="sinusoid", rate=6.0, amp=0.6, period=300.0,
                        accumulate=True)
    t, n = 0.0, 0
    while t < 5000.0:
        t += ap.next_interarrival(t, rng)
        n += 1
    assert n / 5000.0 == pytest.approx(6.0, rel=0.05)


def test_off_mode():
    rng = random.Random(0)
    ap = ArrivalProcess(mode="off", rate=1.0)
    assert ap.next_interarrival(0.0, rng) == float("inf")
    assert ap.lambda_t(10.0) == 0.0


def test_validators():
    good = GPUSpec("ok", p_idle=45.0, p_peak=350.0, p_sleep=28.0, tdp=400.0)
    assert validate_gpu_specs([good]) == []
    bad_sleep = GPUSpec("bs", p_idle=10.0, p_peak=100.0, p_sleep=20.0)
    assert any("p_sleep" in m for m in validate_gpu_specs([bad_sleep]))
    over_tdp = GPUSpec("ot", p_idle=100.0, p_peak=400.0, p_sleep=10.0, tdp=300.0)
    assert any("TDP" in m for m in validate_gpu_specs([over_tdp]))
    low = GPUSpec("lo", p_idle=10.0, p_peak=20.0, p_sleep=5.0, tdp=400.0)
    assert any("<=50%" in m for m in validate_gpu_specs([low]))
    neg = GPUSpec("ng", p_idle=-1.0, p_peak=20.0, p_sleep=0.0)
    assert any("negative" in m for m in validate_gpu_specs([neg]))
    bad_alpha = GPUSpec("ba", p_idle=10.0, p_peak=20.0, p_sleep=5.0, alpha=9.0)
    assert any("alpha" in m for m in validate_gpu_specs([bad_alpha]))
    with pytest.raises(ValueError):
        validate_gpu_specs([bad_sleep], strict=True)
