"""Offline training, env-loop trainer, elastic scaling / preemption tests."""
import numpy as np
import pytest
import torch

from distributed_cluster_gpus_amd.configs.paper import paper_scenario
from distributed_cluster_gpus_amd.engine.oracle import OracleEngine
from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess
from distributed_cluster_gpus_amd.rl.agent import CHSACAgent, CHSACAgentConfig
from distributed_cluster_gpus_amd.rl.env_loop import EnvLoopTrainer
from distributed_cluster_gpus_amd.rl.offline import train_offline
from distributed_cluster_gpus_amd.rl.replay import ReplayRing, save_offline_npz


def _toy_dataset(tmp_path, n=300, obs=11, n_dc=4, n_g=3):
    rng = np.random.default_rng(0)
    data = {
        "s": rng.normal(size=(n, obs)).astype(np.float32),
        "s_next": rng.normal(size=(n, obs)).astype(np.float32),
        "a_dc": rng.integers(0, n_dc, n).astype(np.int64),
        "a_g": rng.integers(0, n_g, n).astype(np.int64),
        "r": rng.normal(size=n).astype(np.float32),
        "done": np.ones(n, np.float32),
        "mask_dc": np.ones((n, n_dc), np.bool_),
        "mask_g": np.ones((n, n_g), np.bool_),
        "costs/latency_p99": (rng.random(n) * 100).astype(np.float32),
    }
    p = str(tmp_path / "ds.npz")
    save_offline_npz(p, data)
    return p


def test_offline_training(tmp_path):
    torch.manual_seed(0)
    p = _toy_dataset(tmp_path)
    agent, stats = train_offline(p, epochs=2, batch_size=64,
                                 constraints={"latency_p99": 50.0})
    assert len(stats) == 2 * (300 // 64)
    assert all(np.isfinite(s["loss_critic"]) for s in stats)
    assert agent.cfg.n_dc == 4 and agent.cfg.n_g_choices == 3


class _ToyEnv:
    def __init__(self):
        self.t = 0

    def get_obs_vector(self):
        return np.full(6, float(self.t % 5), np.float32)

    def get_action_masks(self):
        return np.ones(3, bool), np.ones(2, bool)

    def step(self, a):
        self.t += 1
        r = 1.0 if a["dc"] == 0 else -0.1
        return self.get_obs_vector(), r, False, {"costs": {"c": 1.0}}


def test_env_loop_trainer():
    torch.manual_seed(0)
    agent = CHSACAgent(CHSACAgentConfig(obs_dim=6, n_dc=3, n_g_choices=2,
                                        constraints={"c": 10.0}, device="cpu"))
    ring = ReplayRing(capacity=128, obs_dim=6, n_costs=1, cost_names=["c"],
                      n_dc=3, n_g=2, seed=0)
    trainer = EnvLoopTrainer(agent, ring, batch_size=16, warmup=16)
    env = _ToyEnv()
    stats = {}
    for _ in range(40):
        stats = trainer.step_env_and_learn(env)
    assert ring.size == 40
    assert "loss_critic" in stats  # learning started after warmup


def test_elastic_scaling_preempts_and_resumes(tmp_path):
    """Elastic scaling (chsac_af only): on a training-job completion with >1
    training jobs running, all training jobs are preempted, the agent picks
    new allocations, and jobs resume with preserved progress
    (reference :340-409, :498-534; our CLI can actually enable it —
    the reference's --elastic-scaling flag cannot be True, Appendix A.2)."""
    torch.manual_seed(0)
    sc = paper_scenario()
    inf = ArrivalProcess(mode="off", rate=0.0)
    trn = ArrivalProcess(mode="poisson", rate=0.5)
    eng = OracleEngine(sc, inf, trn, algo="chsac_af", duration=1500.0,
                       log_interval=10.0, out_dir=str(tmp_path / "el"),
                       seed=5, elastic_scaling=True, rl_device="cpu",
                       rl_batch=16, rl_warmup=10**9)  # no training, just act
    eng.run()
    preempts = sum(1 for dc in eng.dcs.values() for _ in dc.preempted_jobs)
    total_preempt_count = 0
    import csv
    import os
    with open(os.path.join(str(tmp_path / "el"), "job_log.csv")) as f:
        for r in csv.DictReader(f):
            total_preempt_count += int(r["preempt_count"])
    # the workload (multiple concurrent training jobs) must have triggered
    # preemption+resume cycles, visible in job_log's preempt_count column
    assert total_preempt_count > 0
    # no job stranded forever in preempted state at end-of-sim beyond those
    # whose resume legitimately deferred (they get re-queued in our fix)
    assert preempts == 0


def test_preempt_resume_preserves_progress(tmp_path):
    """Direct unit check of the preemption checkpoint: units_done at resume
    equals units_done at preempt (job-level checkpoint semantics)."""
    torch.manual_seed(0)
    sc = paper_scenario()
    inf = ArrivalProcess(mode="off", rate=0.0)
    trn = ArrivalProcess(mode="off", rate=0.0)
    eng = OracleEngine(sc, inf, trn, algo="chsac_af", duration=100.0,
                       log_interval=10.0, out_dir=None, seed=1,
                       elastic_scaling=True, rl_device="cpu",
                       rl_batch=8, rl_warmup=10**9)
    from distributed_cluster_gpus_amd.models.cluster import JobState
    dc = eng.dcs["us-west"]
    job = JobState(jid=999, ingress="gw-us-west", jtype="training", size=1000.0,
                   arrival_time=0.0)
    eng.now = 10.0
    eng._start_with_nf(dc, job, 4, 0.8)
    eng.now = 50.0
    eng._preempt_job(dc, job, "test")
    assert dc.free_gpus == dc.total_gpus
    pre = dc.preempted_jobs[0]
    done_at_preempt = pre.ckpt["units_done"]
    assert done_at_preempt > 0
    eng.now = 60.0
    ok = eng._resume_preempted(dc, pre, 2, 0.6)
    assert ok
    assert job.units_done == done_at_preempt
    assert job.gpus_assigned == 2 and job.f_used == 0.6
    assert job.total_preempt_time == pytest.approx(10.0)
    assert dc.busy_gpus == 2
