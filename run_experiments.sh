#!/usr/bin/env bash
# Canonical experiment matrix (capability parity: reference run.sh /
# multi_dc.bat — duration 604800 s, inference off, training 0.02/s,
# log every 20 s; note the reference scripts pass legacy algo names its own
# CLI rejects — this script uses the real names).
#
# Usage: ./run_experiments.sh [out_dir] [engine]
#   engine: oracle | native | batched   (default native; chsac_af always
#   runs through the torch path)
set -euo pipefail

OUT=${1:-runs_paper}
ENGINE=${2:-native}
DURATION=${DURATION:-604800}
TRN_RATE=${TRN_RATE:-0.02}
LOG_INTERVAL=${LOG_INTERVAL:-20}
SEED=${SEED:-123}
HERE=$(cd "$(dirname "$0")" && pwd)

ALGOS=(default_policy joint_nf bandit carbon_cost eco_route cap_greedy chsac_af)

for algo in "${ALGOS[@]}"; do
  eng=$ENGINE
  extra=()
  if [ "$algo" = "chsac_af" ]; then
    eng=oracle
    extra+=(--upgr-device cpu --sla_p99_ms 500)
  fi
  if [ "$algo" = "cap_greedy" ]; then
    extra+=(--power-cap "${POWER_CAP:-60000}")
  fi
  echo "=== $algo ($eng) ==="
  python "$HERE/run_sim.py" \
    --algo "$algo" --engine "$eng" \
    --duration "$DURATION" --log-interval "$LOG_INTERVAL" \
    --inf-mode off --trn-rate "$TRN_RATE" \
    --seed "$SEED" --log-path "$OUT/$algo" --progress False \
    "${extra[@]}"
done

echo "=== comparison report ==="
RUNS=()
for algo in "${ALGOS[@]}"; do
  RUNS+=("$algo=$OUT/$algo")
done
python "$HERE/plot_results.py" --runs "${RUNS[@]}" --out "$OUT/report"
echo "done: $OUT"
