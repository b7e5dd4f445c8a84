#!/usr/bin/env python3
"""CHSAC-AF batched-engine throughput at several replica counts."""
import json
import os
import sys
import time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_cluster_gpus_amd.configs.paper import paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess

for reps in [int(x) for x in (sys.argv[1:] or ["256", "1024", "4096"])]:
    sc = paper_scenario()
    inf = ArrivalProcess(mode="poisson", rate=2.0)
    trn = ArrivalProcess(mode="poisson", rate=0.3)
    eng = BatchedEngine(sc, inf, trn, algo="chsac_af", replicas=reps,
                        duration=300.0, log_interval=5.0, out_dir=None,
                        seed=7, enable_logs=False, rl_warmup=2000,
                        rl_batch=256, rl_train_interval=256,
                        events_per_launch=100000)
    t0 = time.perf_counter()
    st = eng.run()
    wall = time.perf_counter() - t0
    print(json.dumps({
        "replicas": reps, "events": st["events"],
        "events_per_sec": round(st["events"] / wall),
        "jobs": st["jobs_completed"], "rl_updates": eng.rl_updates,
        "rl_updates_per_sec": round(eng.rl_updates / wall, 1),
        "replay": eng.replay.size, "wall_s": round(wall, 2)}), flush=True)
