#!/usr/bin/env bash
# Reproduce the headline results on one MI355X (see BASELINE.md for the
# expected numbers). Usage: bash scripts/reproduce.sh [quick|full]
set -euo pipefail
cd "$(dirname "$0")/.."
MODE="${1:-quick}"

echo "== build (hipcc --offload-arch=gfx950, in-tree) =="
python build.py

echo "== GPU test suite (kernel numerics vs CPU oracle + engine count parity) =="
python -m pytest tests -m gpu -q

echo "== headline (N=17 search + ta014 lb1 proof per step; ~77 ms, ~104 Gnodes/s) =="
python bench.py --gpus 1 --steps 25 --warmup 3

echo "== PFSP ta014 lb1 alone (BASELINE config 3: ~2.1 ms/search) =="
python bench.py --gpus 1 --steps 25 --warmup 5 --problem pfsp --inst 14 --lb lb1

if [ "$MODE" = "full" ]; then
  echo "== N-Queens N=18 (~0.6 s) =="
  python -m gats_amd.cli nqueens --N 18 --tier gpu
  echo "== PFSP ta001-ta020 lb2 sweep, optima proven from ub=1 (~15-20 s total via the CLI) =="
  for i in $(seq 1 20); do
    python -m gats_amd.cli pfsp --inst "$i" --lb lb2 --ub 1 --tier gpu \
        --capacity $((1 << 25)) --stats-file sweep_lb2.dat
  done
  echo "-- sweep results (ta%d lb%d TIER time tree sol optimum) --"
  cat sweep_lb2.dat
fi
