#!/usr/bin/env bash
# Report wrapper (capability parity: reference plot.sh).
# Usage: ./plot.sh <runs_root> [out_dir]
#   <runs_root> contains one subdirectory per algorithm run (each with
#   cluster_log.csv / job_log.csv), as produced by run_experiments.sh.
set -euo pipefail
ROOT=${1:?usage: ./plot.sh <runs_root> [out_dir]}
OUT=${2:-$ROOT/report}
HERE=$(cd "$(dirname "$0")" && pwd)
RUNS=()
for d in "$ROOT"/*/; do
  [ -f "$d/cluster_log.csv" ] || continue
  name=$(basename "$d")
  RUNS+=("$name=$d")
done
[ ${#RUNS[@]} -gt 0 ] || { echo "no runs with cluster_log.csv under $ROOT"; exit 1; }
python "$HERE/plot_results.py" --runs "${RUNS[@]}" --out "$OUT"
first=${RUNS[0]#*=}
python "$HERE/plot_single.py" --run "$first" --out "$OUT/single_$(basename "$first")"
