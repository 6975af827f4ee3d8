#!/bin/bash
# r2 call 22: verify multirank flake fix (full suite) + A/B the shfl
# pre-reduce in the stats epilogues on one box.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c22
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider 2>&1 | tail -4 > gpurun_out/r2c22/pytest.txt
echo "--- bench fused stats OFF (epilogue disabled baseline)"
EDL_BN_STATS_FUSED=0 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c22/bench_off.json
echo "--- bench fused stats ON x2 (with shfl pre-reduce)"
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c22/bench_on1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c22/bench_on2.json
cat gpurun_out/r2c22/*.txt gpurun_out/r2c22/*.json
