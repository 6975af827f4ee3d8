#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c11
echo "== full conv/ops/bn numerics (channels-last buckets + direct 3x3 wgrad) =="
timeout 900 python -m pytest tests/test_gemm_gpu.py tests/test_ops_gpu.py tests/test_bnrelu.py -q -m gpu > gpurun_out/r2c11/pytest.log 2>&1
echo "pytest rc=$?"
echo "== memorization check (trains correctly with new layout) =="
timeout 600 python -m pytest "tests/test_ops_gpu.py::test_full_model_memorizes" -q -m gpu > gpurun_out/r2c11/memorize.log 2>&1
echo "memorize rc=$?"
echo "== bench =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c11/bench.log 2>&1
echo "bench rc=$?"
echo "== distill =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c11/distill.log 2>&1
echo "distill rc=$?"
echo "== trace =="
mkdir -p gpurun_out/r2c11/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c11/prof -o r2s11 -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 10 ) > gpurun_out/r2c11/rocprof.log 2>&1
echo "rocprof rc=$?"
for f in gpurun_out/r2c11/pytest.log gpurun_out/r2c11/bench.log gpurun_out/r2c11/distill.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu.ids; done
