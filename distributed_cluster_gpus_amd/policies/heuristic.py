"""Heuristic in-DC allocator: pick a GPU count and mutate DC-level frequency.

Semantics parity with the reference ``select_gpus_and_set_freq``
(simcore/policy.py:16-41): note it mutates *DC-level* current_freq (jobs then
snapshot it into f_used at start), and always returns at least 1 even when the
DC has no free GPUs — callers gate on free_gpus > 0 first, as the reference's
event handlers do.
"""
from ..models.cluster import DataCenterState, JobState
from ..models.scenario import PolicyParams


def heuristic_allocate(dc: DataCenterState, job: JobState, policy: PolicyParams) -> int:
    free = dc.free_gpus
    g = min(free, policy.max_gpus_per_job) if free > 0 else 0

    if policy.name == "perf_first":
        if job.jtype == "inference":
            dc.current_freq = policy.dvfs_high
            return max(1, g)
        dc.current_freq = max(dc.current_freq,
                              policy.dvfs_high if len(dc.q_inf) > 0 else dc.default_freq)
        return max(1, g)

    if policy.name == "energy_aware":
        if job.jtype == "inference":
            dc.current_freq = policy.dvfs_high
            return max(1, g)
        if policy.train_scale_out_low_freq and free >= 2:
            dc.current_freq = policy.dvfs_low
            g = min(free, policy.max_gpus_per_job)
            return max(1, g)
        dc.current_freq = max(dc.current_freq, policy.dvfs_low)
        return max(1, g)

    raise ValueError(f"Unknown policy name {policy.name!r}")
