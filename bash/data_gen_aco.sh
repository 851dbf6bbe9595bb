#!/bin/bash
# Dataset presets (reference bash/data_gen_aco.sh): train set 200 seeds from
# 100, test set 100 seeds from 500.
set -e
cd "$(dirname "$0")/.."
python data_generation_offloading.py --datapath data/aco_data_ba_200 --gtype ba --size 200 --seed 100
python data_generation_offloading.py --datapath data/aco_data_ba_100 --gtype ba --size 100 --seed 500
