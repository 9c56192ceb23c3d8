#!/bin/bash
# Multi-seed reproducibility sweep (reference `elasticnet/do.sh:1-6`):
# 10 seeds x {hint, nohint}.
DIR="$(cd "$(dirname "$0")" && pwd)"
for seed in 0 1 2 3 4 5 6 7 8 9; do
  python "$DIR/main_sac.py" --seed $seed --episodes 1000 --steps 5 --use_hint > sac_hint_$seed.out
  python "$DIR/main_sac.py" --seed $seed --episodes 1000 --steps 5 > sac_nohint_$seed.out
done
