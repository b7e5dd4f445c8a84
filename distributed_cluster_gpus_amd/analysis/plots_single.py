"""Single-run / per-DC debug report (capability parity: reference
plot_single_algo.py main():271-332 — per-DC queues, utilization, busy GPUs,
energy, the (n, f) operating-point trend, ingress statistics and the
ingress->DC routing heatmap)."""
from typing import List, Optional


from .aggregate import load_run
from .render import emit

SINGLE_FIGURES = [
    "dc_queues", "dc_util", "dc_busy", "dc_energy", "dc_power",
    "nf_trend", "ingress_stats", "routing_heatmap",
]


def single_algo_report(run_dir: str, out_dir: str,
                       dc_labels: Optional[dict] = None,
                       gw_labels: Optional[dict] = None) -> List[str]:
    """Emit every per-DC figure for one run.  dc_labels/gw_labels map raw
    names to display labels (configs.paper.DC_GPUS_LABEL / GW_ALPHABET_LABEL)."""
    cluster, jobs = load_run(run_dir)
    arts = []
    if dc_labels:
        cluster = cluster.copy()
        cluster["dc"] = cluster["dc"].map(lambda d: f"{d} ({dc_labels.get(d, '')})")
    q = cluster[["time_s", "dc", "q_inf", "q_train"]].copy()
    q["q_total"] = q["q_inf"] + q["q_train"]
    arts.append(emit(q[["time_s", "q_total", "dc"]], out_dir, "dc_queues",
                     kind="line", x="time_s", y="q_total", hue="dc",
                     title="Queued jobs per DC"))
    arts.append(emit(cluster[["time_s", "util_avg", "dc"]], out_dir, "dc_util",
                     kind="line", x="time_s", y="util_avg", hue="dc",
                     title="Average utilization per DC"))
    arts.append(emit(cluster[["time_s", "busy", "dc"]], out_dir, "dc_busy",
                     kind="line", x="time_s", y="busy", hue="dc",
                     title="Busy GPUs per DC"))
    arts.append(emit(cluster[["time_s", "energy_kJ", "dc"]], out_dir,
                     "dc_energy", kind="line", x="time_s", y="energy_kJ",
                     hue="dc", title="Cumulative energy per DC", ylabel="kJ"))
    arts.append(emit(cluster[["time_s", "power_W", "dc"]], out_dir, "dc_power",
                     kind="line", x="time_s", y="power_W", hue="dc",
                     title="Power per DC", ylabel="W"))

    # (n, f) operating-point trend over completed jobs
    nf = jobs[["finish_s", "n_gpus", "f_used", "type"]].copy()
    nf = nf.sort_values("finish_s")
    arts.append(emit(nf, out_dir, "nf_trend", kind="scatter", x="finish_s",
                     y="f_used", hue="type",
                     title="Per-job frequency over time"))

    # ingress statistics
    ing = jobs.groupby("ingress").agg(
        jobs=("jid", "count"), mean_latency_s=("latency_s", "mean"),
        mean_net_lat_s=("net_lat_s", "mean")).reset_index()
    if gw_labels:
        ing["ingress"] = ing["ingress"].map(
            lambda g: f"{gw_labels.get(g, g)}:{g}")
    arts.append(emit(ing, out_dir, "ingress_stats", kind="bar", x="ingress",
                     y="jobs", title="Jobs per ingress"))

    # routing heatmap: ingress x DC job counts
    heat = jobs.groupby(["ingress", "dc"]).size().unstack(fill_value=0)
    heat.index.name = "ingress"
    arts.append(emit(heat.reset_index(), out_dir, "routing_heatmap",
                     kind="heatmap", title="Routing: jobs per (ingress, DC)"))
    return arts
