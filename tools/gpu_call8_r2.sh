#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c8
echo "== stem revert validation: pytest subset =="
timeout 600 python -m pytest tests/test_gemm_gpu.py -q -m gpu -k "small or stem or s2" > gpurun_out/r2c8/pytest_stem.log 2>&1
echo "stem rc=$?"
timeout 300 python -m pytest tests/test_gpu_extras.py::test_ctr_wide_and_deep_gpu_step -q -m gpu > gpurun_out/r2c8/pytest_ctr.log 2>&1
echo "ctr rc=$?"
echo "== bench =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c8/bench.log 2>&1
echo "bench rc=$?"
echo "== bench warmup 5 (find-phase check with stem fwd/dgrad in-repo) =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/r2c8/bench_w5.log 2>&1
echo "bench5 rc=$?"
echo "== distill =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c8/distill.log 2>&1
echo "distill rc=$?"
for f in gpurun_out/r2c8/*.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu.ids; done
