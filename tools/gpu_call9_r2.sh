#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c9
echo "== conv numerics (splitk paths) =="
timeout 900 python -m pytest tests/test_gemm_gpu.py tests/test_ops_gpu.py -q -m gpu > gpurun_out/r2c9/pytest.log 2>&1
echo "pytest rc=$?"
echo "== bench =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 10 > gpurun_out/r2c9/bench.log 2>&1
echo "bench rc=$?"
echo "== distill =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c9/distill.log 2>&1
echo "distill rc=$?"
echo "== trace =="
mkdir -p gpurun_out/r2c9/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c9/prof -o r2s9 -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 10 ) > gpurun_out/r2c9/rocprof.log 2>&1
echo "rocprof rc=$?"
for f in gpurun_out/r2c9/pytest.log gpurun_out/r2c9/bench.log gpurun_out/r2c9/distill.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu.ids; done
