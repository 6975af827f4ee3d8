#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c10
echo "== numerics after alignment fix =="
timeout 900 python -m pytest tests/test_bnrelu.py tests/test_ops_gpu.py tests/test_gemm_gpu.py -q -m gpu > gpurun_out/r2c10/pytest.log 2>&1
echo "pytest rc=$?"
echo "== bn bwd microbench =="
timeout 600 python tools/bn_bwd_bench.py --iters 20 > gpurun_out/r2c10/bn_bwd.log 2>&1
echo "bnbwd rc=$?"
echo "== bench =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c10/bench.log 2>&1
echo "bench rc=$?"
echo "== distill =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c10/distill.log 2>&1
echo "distill rc=$?"
echo "== ctr GPU point =="
timeout 300 python tools/bench_ctr.py --steps 200 --warmup 30 --batch 2048 > gpurun_out/r2c10/ctr.log 2>&1
echo "ctr rc=$?"
for f in gpurun_out/r2c10/*.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu.ids; done
grep -h '"bench": "bn_bwd"' gpurun_out/r2c10/bn_bwd.log
