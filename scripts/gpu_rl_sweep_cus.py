#!/usr/bin/env python3
"""Sweep the reserved-CU island size for the overlapped chsac loop:
events/s vs SAC updates/s tradeoff (the advance kernel loses the island;
the train step gains it)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from distributed_cluster_gpus_amd.configs.paper import paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
from distributed_cluster_gpus_amd.models.arrivals import ArrivalProcess

for tgt in [float(x) for x in (sys.argv[1:] or ["220", "150", "0"])]:
    sc = paper_scenario()
    inf = ArrivalProcess(mode="sinusoid", rate=6.0, amp=0.6, period=300.0)
    trn = ArrivalProcess(mode="poisson", rate=0.3)
    eng = BatchedEngine(sc, inf, trn, algo="chsac_af", replicas=4096,
                        duration=1e9, log_interval=20.0, out_dir=None,
                        seed=1, enable_logs=False, rl_warmup=2048,
                        rl_batch=256, rl_train_interval=256,
                        rl_stats_interval=0, events_per_launch=100000,
                        rl_target_updates_per_s=tgt)
    eng.run(max_wall_s=3.0)
    torch.cuda.synchronize()
    ev0 = int(eng.t["ev_count"].sum().item())
    up0 = eng.rl_updates
    t0 = time.perf_counter()
    eng.run(max_wall_s=6.0)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    print(json.dumps({
        "target_ups": tgt,
        "events_per_sec": round((int(eng.t["ev_count"].sum().item()) - ev0) / el),
        "updates_per_sec": round((eng.rl_updates - up0) / el, 1),
        **{k: (round(v, 3) if isinstance(v, float) else v)
           for k, v in eng.timing.items()},
    }), flush=True)
    del eng
    torch.cuda.empty_cache()
