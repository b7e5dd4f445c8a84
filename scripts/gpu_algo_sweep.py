#!/usr/bin/env python3
"""Whole-workload GPU throughput for every non-RL algorithm (profiles)."""
import json
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from gpu_perf import run_one  # noqa: E402

for algo in ("default_policy", "joint_nf", "bandit", "carbon_cost",
             "eco_route", "debug", "cap_greedy", "cap_uniform"):
    kw = {}
    r = run_one(8192, duration=600.0, algo=algo)
    print(json.dumps(r), flush=True)
