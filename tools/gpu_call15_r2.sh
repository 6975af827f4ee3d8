#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c15
echo "== TN TR numerics (wgrad suites) =="
timeout 900 python -m pytest tests/test_gemm_gpu.py -q -m gpu > gpurun_out/r2c15/pytest_gemm.log 2>&1
echo "gemm rc=$?"
timeout 600 python -m pytest tests/test_ops_gpu.py -q -m gpu -k "maxpool or direct_grad or memorizes" > gpurun_out/r2c15/pytest_mp.log 2>&1
echo "mp rc=$?"
echo "== tn A/B timings =="
timeout 300 python tools/tn_pmc_probe.py > gpurun_out/r2c15/tn_tr.log 2>&1
echo "tr rc=$?"
EDL_TN_TR=0 timeout 300 python tools/tn_pmc_probe.py > gpurun_out/r2c15/tn_u16.log 2>&1
echo "u16 rc=$?"
echo "== bench A/B =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c15/bench_tr.log 2>&1
echo "bench_tr rc=$?"
EDL_TN_TR=0 timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c15/bench_u16.log 2>&1
echo "bench_u16 rc=$?"
for f in gpurun_out/r2c15/*.log; do echo "--- $f"; tail -5 "$f" | grep -v amdgpu; done
