#!/bin/bash
# r2 call 24: A/B the gemm_bt LDS-bounce store path.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c24
# numerics under the new path
EDL_BT_STORE_LDS=1 timeout 600 python -m pytest tests/test_gemm_gpu.py tests/test_bnrelu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c24/pytest_ldsb.txt
# per-shape micro A/B
timeout 420 python tools/bt_store_bench.py > gpurun_out/r2c24/bt_store_ab.json 2>gpurun_out/r2c24/bt_store_ab.err
# end-to-end A/B same box
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c24/bench_off.json
EDL_BT_STORE_LDS=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c24/bench_on.json
EDL_BT_STORE_LDS=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c24/bench_on2.json
cat gpurun_out/r2c24/pytest_ldsb.txt gpurun_out/r2c24/bt_store_ab.json gpurun_out/r2c24/bench_*.json
