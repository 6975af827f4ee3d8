#!/bin/bash
# BASELINE config 1: linear-regression plumbing check on CPU (gloo, world 2).
set -e
# run from anywhere: the repo root is importable
export PYTHONPATH="$(cd "$(dirname "$0")/.." && pwd)${PYTHONPATH:+:$PYTHONPATH}"
CUDA_VISIBLE_DEVICES="" EDL_NPROC_PER_NODE=2 \
python -m edl_amd.launch --standalone --nodes_range 1:1 --nproc_per_node 2 \
    --job_id fit_a_line -- -m edl_amd.train.train_simple --model fit_a_line
