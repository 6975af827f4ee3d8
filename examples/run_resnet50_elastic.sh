#!/bin/bash
# Elastic ResNet50_vd training on one node, 1..8 GPUs (BASELINE configs 2-3).
# Start N of these agents (e.g. in tmux panes); kill/add agents to resize.
set -e
# run from anywhere: the repo root is importable
export PYTHONPATH="$(cd "$(dirname "$0")/.." && pwd)${PYTHONPATH:+:$PYTHONPATH}"
STORE=${STORE:-127.0.0.1:2379}
python -m edl_amd.coord.server --port "${STORE##*:}" &
sleep 1
python -m edl_amd.launch --store_endpoints "$STORE" --nodes_range 1:8 \
    --job_id resnet50_vd -- -m edl_amd.train.train_resnet \
    --model resnet50_vd --batch_size 32 --num_epochs 90 \
    --steps_per_epoch 500 --checkpoint ./ckpt_resnet50
