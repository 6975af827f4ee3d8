#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c26
timeout 600 python -m pytest tests/test_bnrelu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c26/pytest.txt
timeout 300 python tools/bn_bwd_bench.py --iters 30 > gpurun_out/r2c26/bn_bwd_base.json 2>&1 || true
EDL_BN_BWD_FUSED=1 timeout 300 python tools/bn_bwd_bench.py --iters 30 > gpurun_out/r2c26/bn_bwd_fused.json 2>&1 || true
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c26/bench_base.json
EDL_BN_BWD_FUSED=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c26/bench_fused.json
EDL_BN_BWD_FUSED=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c26/bench_fused2.json
cat gpurun_out/r2c26/pytest.txt
grep -h total_us gpurun_out/r2c26/bn_bwd_*.json | tail -2
cat gpurun_out/r2c26/bench_*.json
