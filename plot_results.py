#!/usr/bin/env python3
"""Comparison-report CLI (capability parity: reference plot_sim_result.py).

Usage:
  python plot_results.py --runs default_policy=/path/run1 joint_nf=/path/run2 \
      --out report/
Each figure is emitted as CSV data (always) and PNG (when matplotlib is
installed)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--runs", nargs="+", required=True,
                   help="label=run_dir pairs (run_dir contains the CSV logs)")
    p.add_argument("--out", type=str, default="report")
    args = p.parse_args(argv)
    runs = {}
    for spec in args.runs:
        if "=" in spec:
            label, rd = spec.split("=", 1)
        else:
            label, rd = os.path.basename(os.path.normpath(spec)), spec
        runs[label] = rd
    from distributed_cluster_gpus_amd.analysis.plots import comparison_report
    arts = comparison_report(runs, args.out)
    print(f"wrote {len(arts)} artifacts to {args.out}")
    return arts


if __name__ == "__main__":
    main()
