#!/bin/bash
# r2 call 36: distill pipeline sustained soak + elastic resize refresh.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c36
timeout 700 python tools/bench_distill.py --steps 120 --warmup 5 2>&1 | tail -2 > gpurun_out/r2c36/distill_long.txt
timeout 600 python tools/bench_elastic.py --share-gpu0 --rejoin 2>&1 | tail -5 > gpurun_out/r2c36/elastic.txt
timeout 300 python tools/bench_distill.py --steps 20 --warmup 5 2>&1 | tail -1 > gpurun_out/r2c36/distill_short.json
cat gpurun_out/r2c36/distill_long.txt gpurun_out/r2c36/elastic.txt gpurun_out/r2c36/distill_short.json
