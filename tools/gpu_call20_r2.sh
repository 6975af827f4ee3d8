#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c20
echo "== DEEP numerics =="
EDL_TN_DEEP=1 timeout 600 python -m pytest tests/test_gemm_gpu.py -q -m gpu -k "tn or wgrad or conv1x1 or backward" > gpurun_out/r2c20/pytest_deep.log 2>&1
echo "deep pytest rc=$?"
echo "== tn probe A/B/C =="
timeout 300 python tools/tn_pmc_probe.py > gpurun_out/r2c20/tn_base.log 2>&1
EDL_TN_DEEP=1 timeout 300 python tools/tn_pmc_probe.py > gpurun_out/r2c20/tn_deep.log 2>&1
EDL_TN_DEEP=1 EDL_TN_TR=0 timeout 300 python tools/tn_pmc_probe.py > gpurun_out/r2c20/tn_deep_u16.log 2>&1
echo probes done
echo "== bench A/B =="
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c20/bench_base.log 2>&1
EDL_TN_DEEP=1 timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c20/bench_deep.log 2>&1
for f in gpurun_out/r2c20/*.log; do echo "--- $f"; tail -5 "$f" | grep -vE "amdgpu"; done
