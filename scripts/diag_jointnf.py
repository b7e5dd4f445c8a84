import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from distributed_cluster_gpus_amd.configs.paper import build_arrivals, paper_scenario
from distributed_cluster_gpus_amd.engine.batched import BatchedEngine
from distributed_cluster_gpus_amd.engine.oracle import OracleEngine

duration = 90.0
sc = paper_scenario()
inf, trn = build_arrivals()
rec = []
OracleEngine(sc, inf, trn, algo="joint_nf", duration=duration, log_interval=5.0,
             out_dir="/tmp/o", seed=123, arrival_recorder=rec).run()
ing_idx = {n: i for i, n in enumerate(sc.ingress_names)}
dc_idx = {n: i for i, n in enumerate(sc.dc_names)}
NS = sc.n_ing * 2
streams = [[] for _ in range(NS)]
for (tt, ing, jtype, size, dc) in rec:
    streams[ing_idx[ing]*2 + (0 if jtype=="inference" else 1)].append((tt, size, dc))
cap = max(len(x) for x in streams) + 1
times = np.full((1, NS, cap), 1e300)
sizes = np.zeros((1, NS, cap), np.float64)
dcs = np.full((1, NS, cap), -1, np.int8)
for s_id, es in enumerate(streams):
    for k, (tt, size, dc) in enumerate(es):
        times[0,s_id,k]=tt; sizes[0,s_id,k]=size; dcs[0,s_id,k]=dc_idx[dc]
eng = BatchedEngine(paper_scenario(), inf, trn, algo="joint_nf", replicas=1,
                    duration=duration, log_interval=5.0, out_dir="/tmp/g",
                    seed=9, enable_logs=True, arrival_trace=(times, sizes, dcs))
eng.run()
import pandas as pd
co = pd.read_csv("/tmp/o/cluster_log.csv"); cg = pd.read_csv("/tmp/g/cluster_log.csv")
m = co.merge(cg, on=["time_s","dc"], suffixes=("_o","_g"))
m["pd"] = (m.power_W_o - m.power_W_g).abs() / (m.power_W_o.abs()+1e-9)
bad = m.sort_values("pd", ascending=False).head(8)
print(bad[["time_s","dc","power_W_o","power_W_g","run_total_o","run_total_g","busy_o","busy_g","pd"]].to_string())
