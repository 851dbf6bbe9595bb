#!/bin/bash
# Test preset (reference bash/test.sh: BAT800 model, load 0.15, size-100 set).
set -e
cd "$(dirname "$0")/.."
python AdHoc_test.py --datapath data/aco_data_ba_100 \
    --training_set BAT800 --T 1000 --arrival_scale 0.15 --out out "$@"
