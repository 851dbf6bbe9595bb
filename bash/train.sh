#!/bin/bash
# Training preset (reference bash/train.sh: BAT800-style run, load 0.15).
set -e
cd "$(dirname "$0")/.."
python AdHoc_train.py --datapath data/aco_data_ba_200 \
    --training_set BAT800 --T 800 --learning_rate 1e-6 \
    --arrival_scale 0.15 --out out "$@"
