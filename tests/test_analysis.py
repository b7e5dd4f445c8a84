"""Analysis-layer tests: run loading, aggregation, all figure artifacts."""
import os

import pandas as pd
import pytest

from distributed_cluster_gpus_amd.analysis.aggregate import (aggregate_cluster,
                                                             load_run,
                                                             summarize_run)
from distributed_cluster_gpus_amd.analysis.plots import (COMPARISON_FIGURES,
                                                         comparison_report)
from distributed_cluster_gpus_amd.analysis.plots_single import (SINGLE_FIGURES,
                                                                single_algo_report)
from distributed_cluster_gpus_amd.configs.paper import build_arrivals, paper_scenario
from distributed_cluster_gpus_amd.engine.native import NativeEngine


@pytest.fixture(scope="module")
def two_runs(tmp_path_factory):
    base = tmp_path_factory.mktemp("runs")
    dirs = {}
    for algo in ("default_policy", "joint_nf"):
        sc = paper_scenario()
        inf, trn = build_arrivals()
        out = str(base / algo)
        NativeEngine(sc, inf, trn, algo=algo, duration=60.0, log_interval=5.0,
                     out_dir=out, seed=123).run()
        dirs[algo] = out
    return dirs


def test_load_and_aggregate(two_runs):
    cluster, jobs = load_run(two_runs["default_policy"])
    assert set(cluster.columns) >= {"time_s", "dc", "power_W", "energy_kJ"}
    agg = aggregate_cluster(cluster)
    assert (agg["power_W"] > 0).all()
    # cumulative energy is monotonic
    assert agg["energy_kJ"].is_monotonic_increasing
    # 8 DCs per tick folded into one row per tick
    assert len(agg) == cluster["time_s"].nunique()


def test_summarize(two_runs):
    s = summarize_run(two_runs["default_policy"])
    assert s["jobs_completed"] > 0
    assert s["total_energy_kJ"] > 0
    assert s["mean_inf_latency_s"] > 0
    assert s["energy_per_unit_J"] > 0


def test_comparison_report_artifacts(two_runs, tmp_path):
    out = str(tmp_path / "report")
    arts = comparison_report(two_runs, out)
    produced = {os.path.splitext(os.path.basename(a))[0] for a in arts}
    for fig in COMPARISON_FIGURES:
        assert os.path.exists(os.path.join(out, f"{fig}.csv")), fig
    assert os.path.exists(os.path.join(out, "summary.csv"))
    summary = pd.read_csv(os.path.join(out, "summary.csv"))
    assert set(summary["algo"]) == {"default_policy", "joint_nf"}
    # the reference's headline claim shows up in the data: joint_nf uses less
    # energy than default_policy
    e = summary.set_index("algo")["total_energy_kJ"]
    assert e["joint_nf"] < e["default_policy"]


def test_single_report_artifacts(two_runs, tmp_path):
    from distributed_cluster_gpus_amd.configs.paper import (DC_GPUS_LABEL,
                                                            GW_ALPHABET_LABEL)
    out = str(tmp_path / "single")
    single_algo_report(two_runs["default_policy"], out,
                       dc_labels=DC_GPUS_LABEL, gw_labels=GW_ALPHABET_LABEL)
    for fig in SINGLE_FIGURES:
        assert os.path.exists(os.path.join(out, f"{fig}.csv")), fig
    heat = pd.read_csv(os.path.join(out, "routing_heatmap.csv"))
    assert len(heat) == 8  # 8 ingresses
    assert len(heat.columns) == 9  # ingress + 8 DCs


def test_plot_cli(two_runs, tmp_path):
    import plot_results
    import plot_single
    out = str(tmp_path / "cli_report")
    arts = plot_results.main(["--runs"] +
                             [f"{k}={v}" for k, v in two_runs.items()] +
                             ["--out", out])
    assert len(arts) >= 13
    out2 = str(tmp_path / "cli_single")
    arts2 = plot_single.main(["--run", two_runs["default_policy"],
                              "--out", out2])
    assert len(arts2) == len(SINGLE_FIGURES)


def test_plot_sh_wrapper(two_runs, tmp_path):
    """plot.sh discovers run subdirectories and emits the full report."""
    import shutil
    import subprocess
    import sys
    root = str(tmp_path / "runs_root")
    os.makedirs(root)
    for name, d in two_runs.items():
        shutil.copytree(d, os.path.join(root, name))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(["bash", os.path.join(repo, "plot.sh"), root],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    assert os.path.exists(os.path.join(root, "report", "summary.csv"))


def test_violin_and_boxen_render(tmp_path):
    """The violin / letter-value latency views render as PNGs (matplotlib is
    present here) and always emit their CSV data artifacts."""
    import numpy as np
    import pandas as pd
    from distributed_cluster_gpus_amd.analysis.render import emit, have_mpl
    rng = np.random.default_rng(0)
    df = pd.DataFrame({
        "latency_s": np.concatenate([rng.exponential(0.01, 400),
                                     rng.exponential(0.05, 400)]),
        "algo": ["a"] * 400 + ["b"] * 400,
    })
    for kind in ("violin", "boxen"):
        path = emit(df, str(tmp_path), f"lat_{kind}", kind=kind,
                    y="latency_s", hue="algo", title=kind)
        assert os.path.exists(os.path.join(str(tmp_path), f"lat_{kind}.csv"))
        if have_mpl():
            assert os.path.exists(os.path.join(str(tmp_path),
                                               f"lat_{kind}.png"))
