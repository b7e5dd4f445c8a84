"""Heuristic in-DC allocator: GPU count + DC-level DVFS mutation.

Behavioral spec (reference simcore/policy.py:16-41, re-derived as a pure
decision table + a thin applier): the decision depends only on
(policy, jobtype, free, inference-queue depth, current/default f); it mutates
*DC-level* current_freq (jobs snapshot it into f_used at start) and always
returns at least 1 even with no free GPUs — callers gate on free_gpus > 0
first, as the reference's event handlers do.  The device-side twin is
``heuristic_alloc`` in ops/csrc/hip/replica_engine.hip.
"""
from typing import Tuple

from ..models.cluster import DataCenterState, JobState
from ..models.scenario import PolicyParams


def heuristic_decision(policy: PolicyParams, jtype: str, free: int,
                       q_inf_len: int, current_freq: float,
                       default_freq: float) -> Tuple[int, float]:
    """Pure form: (gpu_count, new_dc_frequency) with no state touched."""
    g = min(free, policy.max_gpus_per_job) if free > 0 else 0

    if policy.name == "perf_first":
        if jtype == "inference":
            return max(1, g), policy.dvfs_high
        floor = policy.dvfs_high if q_inf_len > 0 else default_freq
        return max(1, g), max(current_freq, floor)

    if policy.name == "energy_aware":
        if jtype == "inference":
            return max(1, g), policy.dvfs_high
        if policy.train_scale_out_low_freq and free >= 2:
            return max(1, min(free, policy.max_gpus_per_job)), policy.dvfs_low
        return max(1, g), max(current_freq, policy.dvfs_low)

    raise ValueError(f"Unknown policy name {policy.name!r}")


def heuristic_allocate(dc: DataCenterState, job: JobState,
                       policy: PolicyParams) -> int:
    """Apply the decision to the DC (mutates current_freq) and return g."""
    g, new_f = heuristic_decision(policy, job.jtype, dc.free_gpus,
                                  len(dc.q_inf), dc.current_freq,
                                  dc.default_freq)
    dc.current_freq = new_f
    return g
