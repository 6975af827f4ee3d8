#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c12
echo "== numerics: maxpool + repack + full conv/bn =="
timeout 900 python -m pytest tests/test_gemm_gpu.py tests/test_ops_gpu.py tests/test_bnrelu.py -q -m gpu > gpurun_out/r2c12/pytest.log 2>&1
echo "pytest rc=$?"
echo "== bench x2 =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c12/bench1.log 2>&1
echo "bench1 rc=$?"
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c12/bench2.log 2>&1
echo "bench2 rc=$?"
echo "== distill x3 (median vs 656) =="
for i in 1 2 3; do
  timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c12/distill_$i.log 2>&1
  echo "distill$i rc=$?"
done
for f in gpurun_out/r2c12/pytest.log gpurun_out/r2c12/bench*.log gpurun_out/r2c12/distill_*.log; do echo "--- $f"; tail -2 "$f" | grep -v amdgpu.ids; done
