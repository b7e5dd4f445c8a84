#!/usr/bin/env python3
"""Whole-workload GPU perf measurement: run the batched engine to completion
on the BASELINE config (default_policy, 1200 s, paper topology) and report
aggregate events/sec for several replica counts."""
import json
import sys

import torch
import time
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


from distributed_cluster_gpus_amd.configs.paper import build_arrivals, paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine


def run_one(replicas, duration=1200.0, algo="default_policy", qcap=24576, subwave=64):
    sc = paper_scenario()
    inf, trn = build_arrivals()
    eng = BatchedEngine(sc, inf, trn, algo=algo, replicas=replicas,
                        duration=duration, log_interval=5.0, out_dir=None,
                        seed=123, enable_logs=False, qcap=qcap,
                        events_per_launch=100000, subwave=subwave)
    t0 = time.perf_counter()
    st = eng.run()
    wall = time.perf_counter() - t0
    return {
        "replicas": replicas, "algo": algo, "duration": duration,
        "events": st["events"], "wall_s": round(wall, 3),
        "events_per_sec": round(st["events"] / wall),
        "events_per_replica": round(st["events"] / replicas),
        "jobs": st["jobs_completed"],
        "jobs_per_replica": round(st["jobs_completed"] / replicas, 1),
        "mean_inf_latency_ms": round(st["mean_inf_latency_s"] * 1000, 3),
        "energy_kj_per_replica": round(st["mean_energy_j_per_replica"] / 1000, 1),
    }


if __name__ == "__main__":
    args = sys.argv[1:]
    sw = 64
    if args and args[0] == "--mw":
        sw = 8
        args = args[1:]
    reps = [int(x) for x in (args or ["512", "2048", "8192"])]
    for r in reps:
        print(json.dumps(run_one(r, subwave=sw)), flush=True)
        import gc
        gc.collect()
        torch.cuda.empty_cache()
