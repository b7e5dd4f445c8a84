"""Monte-Carlo population statistics from a batched-engine run.

The scalar reference simulates ONE trajectory per run; the batched MI355X
engine advances thousands of replicas, so every output becomes a
DISTRIBUTION.  This module turns the engine's per-replica metric tensors into
a population report: mean / std / standard error / percentiles / confidence
intervals for energy, completed jobs and latency — the capability SURVEY §6
names as the rebuild's north star.
"""
from typing import Dict, Optional

import numpy as np
import pandas as pd


def population_frame(engine) -> pd.DataFrame:
    """One row per replica with its headline outcomes."""
    t = engine.t
    jobs = t["jobs_done"].cpu().numpy().astype(np.float64)
    jobs_inf = t["jobs_done_inf"].cpu().numpy().astype(np.float64)
    energy = t["energy_j"].sum(dim=1).cpu().numpy()
    lat = t["sum_lat"].cpu().numpy()
    lat_inf = t["sum_lat_inf"].cpu().numpy()
    with np.errstate(divide="ignore", invalid="ignore"):
        mean_lat = np.where(jobs > 0, lat / jobs, np.nan)
        mean_lat_inf = np.where(jobs_inf > 0, lat_inf / jobs_inf, np.nan)
    return pd.DataFrame({
        "replica": np.arange(len(jobs)) + engine.shard.start,
        "jobs_completed": jobs,
        "jobs_inference": jobs_inf,
        "total_energy_kJ": energy / 1e3,
        "mean_latency_s": mean_lat,
        "mean_inf_latency_ms": mean_lat_inf * 1e3,
        "events": t["ev_count"].cpu().numpy().astype(np.float64),
    })


def population_report(engine, out_dir: Optional[str] = None,
                      confidence: float = 0.95) -> Dict:
    """Population statistics dict (+ CSV/figure artifacts when out_dir set).

    Returns, per metric: mean, std (across replicas), standard error of the
    mean, the normal-approximation confidence interval, and percentiles —
    i.e. Monte-Carlo error bars on every simulator output.
    """
    df = population_frame(engine)
    z = {0.90: 1.6449, 0.95: 1.9600, 0.99: 2.5758}.get(round(confidence, 2),
                                                       1.9600)
    metrics = ["jobs_completed", "total_energy_kJ", "mean_latency_s",
               "mean_inf_latency_ms"]
    report: Dict = {"replicas": int(len(df)), "confidence": confidence}
    rows = []
    for m in metrics:
        v = df[m].dropna().values
        if len(v) == 0:
            continue
        mean, std = float(np.mean(v)), float(np.std(v, ddof=1)) if len(v) > 1 else 0.0
        se = std / np.sqrt(len(v)) if len(v) > 1 else 0.0
        entry = {
            "metric": m, "mean": mean, "std": std, "stderr": se,
            "ci_lo": mean - z * se, "ci_hi": mean + z * se,
            "p01": float(np.percentile(v, 1)), "p50": float(np.percentile(v, 50)),
            "p99": float(np.percentile(v, 99)),
        }
        rows.append(entry)
        report[m] = entry
    if out_dir is not None:
        import os

        from .render import emit
        os.makedirs(out_dir, exist_ok=True)
        df.to_csv(os.path.join(out_dir, "population.csv"), index=False)
        pd.DataFrame(rows).to_csv(os.path.join(out_dir, "population_stats.csv"),
                                  index=False)
        emit(df[["total_energy_kJ"]], out_dir, "population_energy_hist",
             kind="hist", y="total_energy_kJ",
             title=f"Total energy across {len(df)} replicas")
        emit(df[["jobs_completed"]], out_dir, "population_jobs_hist",
             kind="hist", y="jobs_completed",
             title=f"Completed jobs across {len(df)} replicas")
    return report
