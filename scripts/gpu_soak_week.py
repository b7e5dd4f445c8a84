#!/usr/bin/env python3
"""Canonical week-long experiment (reference run.sh matrix: duration 604800,
inference off, training 0.02/s, log every 20 s) on the batched engine."""
import json
import os
import sys
import time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_cluster_gpus_amd.configs.paper import paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess
from distributed_cluster_gpus_amd.analysis.montecarlo import population_report

sc = paper_scenario()
inf = ArrivalProcess(mode="off", rate=0.0)
trn = ArrivalProcess(mode="poisson", rate=0.02)
eng = BatchedEngine(sc, inf, trn, algo="default_policy", replicas=256,
                    duration=604800.0, log_interval=20.0,
                    out_dir="gpurun_out/week", seed=123, enable_logs=True,
                    events_per_launch=500000)
t0 = time.perf_counter()
st = eng.run()
wall = time.perf_counter() - t0
rep = population_report(eng)
print(json.dumps({
    "sim_days": 7, "replicas": 256, "events": st["events"],
    "events_per_sec": round(st["events"] / wall), "wall_s": round(wall, 2),
    "jobs_per_replica": round(st["jobs_completed"] / 256, 1),
    "energy_GJ_mean": round(rep["total_energy_kJ"]["mean"] / 1e6, 3),
    "energy_GJ_ci": [round(rep["total_energy_kJ"]["ci_lo"] / 1e6, 3),
                     round(rep["total_energy_kJ"]["ci_hi"] / 1e6, 3)],
    "log_rows": sum(1 for _ in open("gpurun_out/week/cluster_log.csv")) - 1,
}))
