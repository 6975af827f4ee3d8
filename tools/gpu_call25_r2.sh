#!/bin/bash
# r2 call 25: A/B fixed LDS-bounce stores + fused BN bwd.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c25
timeout 600 python -m pytest tests/test_bnrelu.py tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c25/pytest.txt
EDL_BT_STORE_LDS=1 timeout 600 python -m pytest tests/test_bnrelu.py tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c25/pytest_ldsb.txt
timeout 300 python tools/bt_store_bench.py > gpurun_out/r2c25/bt_store_ab.json 2>gpurun_out/r2c25/bt_store_ab.err
timeout 300 python tools/bn_bwd_bench.py > gpurun_out/r2c25/bn_bwd_base.json 2>&1 || true
EDL_BN_BWD_FUSED=1 timeout 300 python tools/bn_bwd_bench.py > gpurun_out/r2c25/bn_bwd_fused.json 2>&1 || true
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c25/bench_base.json
EDL_BN_BWD_FUSED=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c25/bench_bnfused.json
EDL_BT_STORE_LDS=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c25/bench_ldsb.json
EDL_BN_BWD_FUSED=1 EDL_BT_STORE_LDS=1 timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c25/bench_both.json
tail -2 gpurun_out/r2c25/pytest.txt gpurun_out/r2c25/pytest_ldsb.txt
cat gpurun_out/r2c25/bt_store_ab.json gpurun_out/r2c25/bn_bwd_*.json gpurun_out/r2c25/bench_*.json
