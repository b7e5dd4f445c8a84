"""distributed_cluster_gpus_amd — an MI355X-native geo-distributed GPU-cluster
energy/scheduling simulation framework.

Capability target: filrg/distributed_cluster_GPUs (reference: a pure-Python,
single-process discrete-event simulator of a multi-datacenter GPU cluster with
DVFS/energy-aware scheduling and a constrained-RL scheduler).  This package is a
ground-up MI355X-first re-architecture, NOT a port:

* ``engine.oracle``  — scalar Python discrete-event engine, semantics-compatible
  with the reference event loop (the correctness oracle).
* ``engine.native``  — C++ scalar DES core (pybind11) with a CPython-compatible
  MT19937 so its logs are bitwise-identical to the oracle's.
* ``engine.batched`` — the MI355X engine: tens of thousands of Monte-Carlo
  replicas advance in lockstep, one hand-written HIP/CDNA4 (gfx950) kernel per
  step; per-replica state lives in HBM3E as structure-of-arrays.
* ``rl``             — CHSAC-AF (masked hybrid-discrete SAC, quantile critics,
  PID-Lagrangian constraints) on PyTorch-ROCm, data-parallel over RCCL/xGMI.
* ``parallel``       — replica sharding + RCCL collective helpers (one process
  per GPU, ``torch.distributed`` backend "nccl" == RCCL on ROCm).
"""

__version__ = "0.1.0"
