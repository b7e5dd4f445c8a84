from .coeffs import PowerCoeffs, LatencyCoeffs
from .power import gpu_power_w, job_power_w
from .latency import unit_time_s
from .gputypes import GPUSpec, validate_gpu_specs
from .cluster import JobState, PreemptedJobState, DataCenterState, JTYPE_INFERENCE, JTYPE_TRAINING
from .arrivals import ArrivalProcess, sample_job_size
from .wan import WanGraph, IngressSpec, dijkstra_tables
from .scenario import Scenario

__all__ = [
    "PowerCoeffs", "LatencyCoeffs", "gpu_power_w", "job_power_w", "unit_time_s",
    "GPUSpec", "validate_gpu_specs",
    "JobState", "PreemptedJobState", "DataCenterState", "JTYPE_INFERENCE", "JTYPE_TRAINING",
    "ArrivalProcess", "sample_job_size",
    "WanGraph", "IngressSpec", "dijkstra_tables",
    "Scenario",
]
