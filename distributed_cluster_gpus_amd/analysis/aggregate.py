"""Run loading + aggregation for the analysis layer.

Reads only the two CSV logs (capability parity: reference plot_sim_result.py
``load_run``/``aggregate_cluster``, :10-54 — the analysis layer has no other
coupling to the simulator)."""
import os
from typing import Dict

import pandas as pd


def load_run(run_dir: str):
    """Load (cluster_df, job_df) from a run directory."""
    cluster = pd.read_csv(os.path.join(run_dir, "cluster_log.csv"))
    jobs = pd.read_csv(os.path.join(run_dir, "job_log.csv"))
    return cluster, jobs


def aggregate_cluster(cluster: pd.DataFrame) -> pd.DataFrame:
    """Cluster-wide time series: per tick, total power, cumulative energy,
    mean utilization, queue totals."""
    g = cluster.groupby("time_s")
    out = pd.DataFrame({
        "power_W": g["power_W"].sum(),
        "energy_kJ": g["energy_kJ"].sum(),
        "util_inst": g["util_inst"].mean(),
        "util_avg": g["util_avg"].mean(),
        "q_inf": g["q_inf"].sum(),
        "q_train": g["q_train"].sum(),
        "busy": g["busy"].sum(),
        "free": g["free"].sum(),
        "run_total": g["run_total"].sum(),
        "acc_job_unit": g["acc_job_unit"].sum(),
    })
    return out.reset_index()


def summarize_run(run_dir: str) -> Dict:
    """Headline scalars for one run (energy, latency, counts)."""
    cluster, jobs = load_run(run_dir)
    agg = aggregate_cluster(cluster)
    inf = jobs[jobs["type"] == "inference"]
    trn = jobs[jobs["type"] == "training"]
    return {
        "total_energy_kJ": float(agg["energy_kJ"].iloc[-1]) if len(agg) else 0.0,
        "jobs_completed": int(len(jobs)),
        "jobs_inference": int(len(inf)),
        "jobs_training": int(len(trn)),
        "mean_latency_s": float(jobs["latency_s"].mean()) if len(jobs) else 0.0,
        "mean_inf_latency_s": float(inf["latency_s"].mean()) if len(inf) else 0.0,
        "p99_inf_latency_s": float(inf["latency_s"].quantile(0.99)) if len(inf) else 0.0,
        "mean_power_W": float(agg["power_W"].mean()) if len(agg) else 0.0,
        "peak_power_W": float(agg["power_W"].max()) if len(agg) else 0.0,
        "total_job_units": float(jobs["size"].sum()) if len(jobs) else 0.0,
        "energy_per_unit_J": (float(agg["energy_kJ"].iloc[-1]) * 1000.0 /
                              max(1e-9, float(jobs["size"].sum())))
        if len(agg) and len(jobs) else 0.0,
    }
