#!/usr/bin/env bash
# Single-DC controlled-variable debug runs (capability parity: reference
# single_dc_debug.bat): pin n and f on the 128-GPU single-DC topology and
# sweep them, then render the per-DC report.
set -euo pipefail

OUT=${1:-runs_single_dc}
DURATION=${DURATION:-3600}
SEED=${SEED:-123}
HERE=$(cd "$(dirname "$0")" && pwd)

for n in 1 2 4 8; do
  for f in 0.5 0.8 1.0; do
    tag="n${n}_f${f}"
    echo "=== debug $tag ==="
    python "$HERE/run_sim.py" \
      --algo debug --single-dc --engine oracle \
      --num_fixed_gpus "$n" --fixed_freq "$f" \
      --duration "$DURATION" --log-interval 20 \
      --inf-mode poisson --inf-rate 2.0 --trn-mode off \
      --seed "$SEED" --log-path "$OUT/$tag" --progress False
  done
done

python "$HERE/plot_single.py" --run "$OUT/n1_f1.0" --out "$OUT/report_n1_f1.0"
echo "done: $OUT"
