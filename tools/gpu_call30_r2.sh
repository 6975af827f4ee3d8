#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c30
timeout 600 python -m pytest tests/test_gpu_extras.py tests/test_ops_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c30/pytest.txt
EDL_PREFILL_DERIVED=0 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c30/bench_off.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c30/bench_on1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c30/bench_on2.json
timeout 300 python tools/bench_distill.py --steps 20 --warmup 5 2>&1 | tail -1 > gpurun_out/r2c30/distill.json
cat gpurun_out/r2c30/pytest.txt gpurun_out/r2c30/bench_*.json gpurun_out/r2c30/distill.json
