#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c17
echo "== conv-fused BN stats numerics =="
timeout 900 python -m pytest tests/test_bnrelu.py tests/test_ops_gpu.py tests/test_gemm_gpu.py -q -m gpu > gpurun_out/r2c17/pytest.log 2>&1
echo "pytest rc=$?"
echo "== memorization =="
timeout 600 python -m pytest "tests/test_ops_gpu.py::test_full_model_memorizes" -q -m gpu > gpurun_out/r2c17/mem.log 2>&1
echo "mem rc=$?"
echo "== bench A/B =="
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c17/bench_fused.log 2>&1
echo "fused rc=$?"
EDL_BN_STATS_FUSED=0 timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c17/bench_unfused.log 2>&1
echo "unfused rc=$?"
echo "== distill =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c17/distill.log 2>&1
echo "distill rc=$?"
for f in gpurun_out/r2c17/*.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu; done
