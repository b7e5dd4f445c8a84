"""Scalar-engine cluster state: jobs, preemption checkpoints, datacenters.

Capability parity: reference ``Job`` / ``PreemptedJob`` / ``DataCenter``
(simcore/models.py:5-106).  This is the *host* representation used by the
Python oracle; the batched MI355X engine mirrors the same fields as flat
structure-of-arrays device buffers (engine/batched.py, ops/csrc/hip/).
"""
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

JTYPE_INFERENCE = 0
JTYPE_TRAINING = 1

JTYPE_NAMES = ("inference", "training")


@dataclass
class JobState:
    jid: int
    ingress: str
    jtype: str                      # 'inference' | 'training'
    size: float                     # abstract work units
    arrival_time: float
    deadline: Optional[float] = None
    dc_name: Optional[str] = None
    gpus_assigned: int = 0
    start_time: Optional[float] = None
    finish_time: Optional[float] = None
    net_latency_s: float = 0.0
    f_used: float = 0.0             # per-job DVFS frequency
    units_total: float = 0.0
    units_done: float = 0.0
    last_update: float = 0.0        # last progress-integration timestamp
    ev_gen: int = 0                 # event generation (lazy invalidation of stale finish events)
    preemptible: bool = False
    preempt_count: int = 0
    total_preempt_time: float = 0.0
    last_checkpoint: float = 0.0
    # transient per-algorithm annotations (eco hint, RL trace) are attached
    # dynamically by the engines, mirroring the reference's ad-hoc attributes.


@dataclass
class PreemptedJobState:
    """Job-level checkpoint taken at preemption (reference simcore/models.py:30-35)."""
    job: JobState
    preempt_time: float
    reason: str
    ckpt: dict  # {units_done, f_used, gpus_assigned}


@dataclass
class DataCenterState:
    name: str
    gpu_name: str
    p_idle: float
    p_peak: float
    p_sleep: float
    pow_alpha: float
    total_gpus: int
    freq_levels: List[float]
    default_freq: float = 1.0
    power_gating: bool = True

    # runtime
    current_freq: float = field(init=False)
    busy_gpus: int = field(default=0, init=False)
    running_jobs: Dict[int, Tuple[JobState, int]] = field(default_factory=dict, init=False)
    q_inf: List[JobState] = field(default_factory=list, init=False)
    q_train: List[JobState] = field(default_factory=list, init=False)
    energy_joules: float = field(default=0.0, init=False)
    last_energy_time: float = field(default=0.0, init=False)
    util_gpu_time: float = 0.0      # integral of busy_gpus dt  [GPU*s]
    util_last_ts: float = 0.0
    util_begin_ts: float = 0.0
    accumulated_job_unit: float = 0.0
    preempted_jobs: List[PreemptedJobState] = field(default_factory=list, init=False)
    preempt_policy: str = "fifo"

    def __post_init__(self):
        assert self.default_freq in self.freq_levels, "default_freq must be one of freq_levels"
        self.current_freq = self.default_freq

    @property
    def free_gpus(self) -> int:
        return self.total_gpus - self.busy_gpus

    def baseline_power_w(self) -> float:
        """Documented-but-unused baseline model: p_idle + p_peak*f^alpha per active
        GPU, sleep/idle for the rest (reference models.py:82-91; shadowed at runtime
        by the paper model — SURVEY Appendix A.13)."""
        f = self.current_freq
        active = self.busy_gpus
        idle = self.total_gpus - active
        p_active = active * (self.p_idle + self.p_peak * (f ** self.pow_alpha))
        p_idle = idle * (self.p_sleep if self.power_gating else self.p_idle)
        return p_active + p_idle

    def accrue_energy(self, now: float,
                      power_fn: Optional[Callable[["DataCenterState"], float]] = None) -> None:
        """E += P * dt since the last accrual; first call only stamps the time
        (reference models.py:93-106)."""
        if self.last_energy_time == 0.0:
            self.last_energy_time = now
            return
        dt = max(0.0, now - self.last_energy_time)
        p = power_fn(self) if power_fn else self.baseline_power_w()
        self.energy_joules += p * dt
        self.last_energy_time = now
