#!/usr/bin/env python3
"""Per-DC debug report for ONE run (capability parity: reference
plot_single_algo.py).  Usage: python plot_single.py --run <dir> --out report/"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--run", type=str, required=True)
    p.add_argument("--out", type=str, default="report_single")
    args = p.parse_args(argv)
    from distributed_cluster_gpus_amd.analysis.plots_single import single_algo_report
    from distributed_cluster_gpus_amd.configs.paper import (DC_GPUS_LABEL,
                                                            GW_ALPHABET_LABEL)
    arts = single_algo_report(args.run, args.out, dc_labels=DC_GPUS_LABEL,
                              gw_labels=GW_ALPHABET_LABEL)
    print(f"wrote {len(arts)} artifacts to {args.out}")
    return arts


if __name__ == "__main__":
    main()
