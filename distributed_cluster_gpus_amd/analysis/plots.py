"""Algorithm-comparison report: the reference's 11 comparison figure types
(capability parity: plot_sim_result.py main():398-504 — total power over time,
cumulative energy, utilization, queue lengths + CSV table, latency
histogram / box / distribution views, energy-vs-latency scatter, total-energy
bar, throughput, energy-per-unit bar, average-latency bar, completed-jobs
bar), computed from the CSV logs only."""
import os
from typing import Dict, List

import numpy as np
import pandas as pd

from .aggregate import aggregate_cluster, load_run, summarize_run
from .render import emit

COMPARISON_FIGURES = [
    "total_power", "cumulative_energy", "utilization", "queues",
    "latency_hist", "latency_box", "latency_violin", "latency_boxen",
    "latency_dist",
    "energy_vs_latency", "total_energy_bar", "throughput",
    "energy_per_unit_bar", "avg_latency_bar", "completed_jobs_bar",
]


def comparison_report(run_dirs: Dict[str, str], out_dir: str) -> List[str]:
    """run_dirs: {algo_label: run_directory}.  Emits every comparison figure
    (CSV always, PNG when matplotlib is present); returns artifact paths."""
    arts = []
    series = {}
    jobs_all = []
    for label, rd in run_dirs.items():
        cluster, jobs = load_run(rd)
        agg = aggregate_cluster(cluster)
        agg["algo"] = label
        series[label] = agg
        jobs = jobs.copy()
        jobs["algo"] = label
        jobs_all.append(jobs)
    ts = pd.concat(series.values(), ignore_index=True)
    jobs = pd.concat(jobs_all, ignore_index=True)

    arts.append(emit(ts[["time_s", "power_W", "algo"]], out_dir, "total_power",
                     kind="line", x="time_s", y="power_W", hue="algo",
                     title="Total cluster power", ylabel="W"))
    arts.append(emit(ts[["time_s", "energy_kJ", "algo"]], out_dir,
                     "cumulative_energy", kind="line", x="time_s",
                     y="energy_kJ", hue="algo",
                     title="Cumulative energy", ylabel="kJ"))
    arts.append(emit(ts[["time_s", "util_avg", "algo"]], out_dir, "utilization",
                     kind="line", x="time_s", y="util_avg", hue="algo",
                     title="Average GPU utilization", ylabel="fraction"))
    q = ts[["time_s", "q_inf", "q_train", "algo"]].copy()
    q["q_total"] = q["q_inf"] + q["q_train"]
    arts.append(emit(q, out_dir, "queues", kind="line", x="time_s",
                     y="q_total", hue="algo", title="Queued jobs", ylabel="jobs"))
    # queue summary table (the reference also writes a CSV table for queues)
    qtab = q.groupby("algo")[["q_inf", "q_train", "q_total"]].agg(["mean", "max"])
    qtab.columns = ["_".join(c) for c in qtab.columns]
    arts.append(emit(qtab.reset_index(), out_dir, "queues_table", kind="bar",
                     x="algo", y="q_total_mean", title="Mean queued jobs"))

    inf = jobs[jobs["type"] == "inference"]
    arts.append(emit(inf[["latency_s", "algo"]], out_dir, "latency_hist",
                     kind="hist", y="latency_s", hue="algo",
                     title="Inference latency histogram"))
    arts.append(emit(inf[["latency_s", "algo"]], out_dir, "latency_box",
                     kind="box", y="latency_s", hue="algo",
                     title="Inference latency distribution (box)"))
    # violin + letter-value (boxen) views, matching the reference's seaborn
    # figures (plot_sim_result.py latency views); raw values are sampled to
    # bound the CSV artifact size
    lat = inf[["latency_s", "algo"]]
    if len(lat) > 50000:
        lat = lat.groupby("algo", group_keys=False).apply(
            lambda g: g.sample(min(len(g), 50000 // max(1, lat["algo"].nunique())),
                               random_state=0))
    arts.append(emit(lat, out_dir, "latency_violin", kind="violin",
                     y="latency_s", hue="algo",
                     title="Inference latency distribution (violin)"))
    arts.append(emit(lat, out_dir, "latency_boxen", kind="boxen",
                     y="latency_s", hue="algo",
                     title="Inference latency distribution (letter-value)"))
    # distribution view (quantile curves as data)
    rows = []
    for label, sub in inf.groupby("algo"):
        qs = np.linspace(0.01, 0.99, 99)
        vals = sub["latency_s"].quantile(qs).values
        rows.append(pd.DataFrame({"quantile": qs, "latency_s": vals,
                                  "algo": label}))
    if rows:
        arts.append(emit(pd.concat(rows, ignore_index=True), out_dir,
                         "latency_dist", kind="line", x="quantile",
                         y="latency_s", hue="algo",
                         title="Inference latency quantiles"))

    summaries = {label: summarize_run(rd) for label, rd in run_dirs.items()}
    sdf = pd.DataFrame.from_dict(summaries, orient="index").reset_index()
    sdf = sdf.rename(columns={"index": "algo"})
    arts.append(emit(sdf[["algo", "mean_inf_latency_s", "total_energy_kJ"]],
                     out_dir, "energy_vs_latency", kind="scatter",
                     x="mean_inf_latency_s", y="total_energy_kJ", hue="algo",
                     title="Energy vs latency trade-off"))
    arts.append(emit(sdf[["algo", "total_energy_kJ"]], out_dir,
                     "total_energy_bar", kind="bar", x="algo",
                     y="total_energy_kJ", title="Total energy", ylabel="kJ"))
    arts.append(emit(ts[["time_s", "acc_job_unit", "algo"]], out_dir,
                     "throughput", kind="line", x="time_s", y="acc_job_unit",
                     hue="algo", title="Accumulated job units"))
    arts.append(emit(sdf[["algo", "energy_per_unit_J"]], out_dir,
                     "energy_per_unit_bar", kind="bar", x="algo",
                     y="energy_per_unit_J", title="Energy per job unit",
                     ylabel="J/unit"))
    arts.append(emit(sdf[["algo", "mean_inf_latency_s"]], out_dir,
                     "avg_latency_bar", kind="bar", x="algo",
                     y="mean_inf_latency_s", title="Mean inference latency",
                     ylabel="s"))
    arts.append(emit(sdf[["algo", "jobs_completed"]], out_dir,
                     "completed_jobs_bar", kind="bar", x="algo",
                     y="jobs_completed", title="Completed jobs"))
    # full summary table
    sdf.to_csv(os.path.join(out_dir, "summary.csv"), index=False)
    arts.append(os.path.join(out_dir, "summary.csv"))
    return arts
