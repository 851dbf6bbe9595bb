#!/bin/bash
# MI355X-native fast trainer preset (one GPU; use torchrun for DP).
set -e
cd "$(dirname "$0")/.."
python -m multihop_offload_amd.harness.train_batched \
    --steps 50000 --batch 1536 --sizes 20,30,40,50,60,70,80,90,100,110 \
    --distinct 256 --explore_decay 0.9998 --lr_decay_at 35000 \
    --eval_every 1000 "$@"
