"""CLI surface + output-dir rule + end-to-end runs through run_sim.main."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import run_sim  # noqa: E402


def test_flag_surface_defaults():
    a = run_sim.parse_args([])
    assert a.duration == 180.0 and a.policy == "energy_aware"
    assert a.log_interval == 5.0 and a.seed == 123
    assert a.inf_mode == "sinusoid" and a.inf_rate == 6.0 and a.inf_amp == 0.6
    assert a.inf_period == 300.0 and a.trn_mode == "poisson" and a.trn_rate == 0.3
    assert a.algo == "default_policy" and a.power_cap == 0.0
    assert a.control_interval == 5.0 and a.eco_objective == "energy"
    assert a.num_fixed_gpus == 1 and a.fixed_freq is None
    assert a.upgr_buffer == 200_000 and a.upgr_batch == 256
    assert a.upgr_warmup == 1_000 and a.upgr_device == "cuda"
    assert a.sla_p99_ms == 500.0 and a.energy_budget_j is None
    assert a.elastic_scaling is False


def test_algo_choices_match_reference():
    for algo in ("default_policy", "cap_uniform", "cap_greedy", "joint_nf",
                 "bandit", "carbon_cost", "eco_route", "chsac_af", "debug"):
        assert run_sim.parse_args(["--algo", algo]).algo == algo
    with pytest.raises(SystemExit):
        run_sim.parse_args(["--algo", "nope"])


def test_out_dir_rule():
    # bare name -> algo subdir; path with separator -> as-is
    assert run_sim.resolve_out_dir("logs", "bandit") == os.path.join("logs", "bandit")
    assert run_sim.resolve_out_dir("/tmp/x", "bandit") == os.path.normpath("/tmp/x")
    assert run_sim.resolve_out_dir(None, "bandit") == os.getcwd()


def test_end_to_end_main(tmp_path):
    out = str(tmp_path / "run")
    stats = run_sim.main(["--algo", "default_policy", "--duration", "30",
                          "--log-path", out, "--progress", "False"])
    assert stats["events"] > 0
    assert os.path.exists(os.path.join(out, "cluster_log.csv"))
    assert os.path.exists(os.path.join(out, "job_log.csv"))
    assert os.path.exists(os.path.join(out, "project.log"))


def test_end_to_end_single_dc_debug(tmp_path):
    out = str(tmp_path / "run")
    stats = run_sim.main(["--algo", "debug", "--single-dc", "--duration", "30",
                          "--num_fixed_gpus", "2", "--fixed_freq", "0.5",
                          "--inf-mode", "poisson", "--inf-rate", "2.0",
                          "--log-path", out, "--progress", "False"])
    assert stats["jobs_completed"] > 0


def test_end_to_end_chsac_cpu(tmp_path):
    out = str(tmp_path / "run")
    ckpt = str(tmp_path / "agent.pt")
    stats = run_sim.main(["--algo", "chsac_af", "--duration", "60",
                          "--inf-mode", "off", "--trn-rate", "0.5",
                          "--upgr-warmup", "10", "--upgr-batch", "8",
                          "--upgr-device", "cpu",
                          "--rl-checkpoint", ckpt,
                          "--log-path", out, "--progress", "False"])
    assert os.path.exists(ckpt)
    assert stats["events"] > 0


def test_subprocess_invocation(tmp_path):
    """The CLI must work as a standalone script (reference run.sh usage)."""
    out = str(tmp_path / "sp")
    r = subprocess.run([sys.executable, os.path.join(REPO, "run_sim.py"),
                        "--algo", "default_policy", "--duration", "10",
                        "--log-path", out, "--progress", "False"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    assert "Done. (default_policy)" in r.stdout
