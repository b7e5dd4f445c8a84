#!/usr/bin/env python3
"""CHSAC-AF on the paper's canonical RL configuration (README.md:268 /
run.sh matrix: inference off, training 0.02/s, SLA 500 ms) — the BASELINE
config-4 measurement: RL-in-the-loop events/s + SAC updates/s on the real
workload (not the synthetic-rate loop bench)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from distributed_cluster_gpus_amd.configs.paper import paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess

duration = float(sys.argv[1]) if len(sys.argv) > 1 else 40000.0
replicas = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
sc = paper_scenario()
inf = ArrivalProcess(mode="off", rate=0.0)
trn = ArrivalProcess(mode="poisson", rate=0.02)
eng = BatchedEngine(sc, inf, trn, algo="chsac_af", replicas=replicas,
                    duration=duration, log_interval=20.0, out_dir=None,
                    seed=123, enable_logs=False, sla_p99_ms=500.0,
                    rl_warmup=200, rl_batch=256, rl_stats_interval=0)
t0 = time.perf_counter()
st = eng.run()
wall = time.perf_counter() - t0
print(json.dumps({
    "config": "BASELINE chsac_af: inf off, trn 0.02/s, sla 500ms",
    "replicas": replicas, "sim_duration_s": duration,
    "events": st["events"], "wall_s": round(wall, 2),
    "events_per_sec": round(st["events"] / wall),
    "rl_updates": eng.rl_updates,
    "updates_per_sec": round(eng.rl_updates / wall, 1),
    "jobs_completed": st["jobs_completed"],
    "replay_size": eng.replay.size,
    "lambda": {k: round(float(v), 4) for k, v in eng.rl.cmdp.lmbda.items()},
}))
