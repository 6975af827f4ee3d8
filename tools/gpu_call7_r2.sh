#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c7
echo "== stem + extras numerics =="
timeout 900 python -m pytest tests/test_gemm_gpu.py tests/test_gpu_extras.py -q -m gpu > gpurun_out/r2c7/pytest.log 2>&1
echo "pytest rc=$?"
echo "== bench with stem kernels (warmup can drop: A/B 15 vs 5) =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c7/bench_w15.log 2>&1
echo "bench15 rc=$?"
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/r2c7/bench_w5.log 2>&1
echo "bench5 rc=$?"
echo "== distill shared =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c7/distill_shared.log 2>&1
echo "distill rc=$?"
echo "== rocprof: steady-state MIOpen check =="
mkdir -p gpurun_out/r2c7/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c7/prof -o r2s7 -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 10 ) > gpurun_out/r2c7/rocprof.log 2>&1
echo "rocprof rc=$?"
echo "== tails =="
for f in gpurun_out/r2c7/pytest.log gpurun_out/r2c7/bench_w15.log gpurun_out/r2c7/bench_w5.log gpurun_out/r2c7/distill_shared.log; do echo "--- $f"; tail -3 "$f" | grep -v amdgpu.ids; done
echo "--- miopen/naive/igemm rows in steady profile:"
grep -iE "igemm|naive|miopen|Cijk|Im2Col|Col2Im" gpurun_out/r2c7/prof/r2s7_kernel_stats.csv | head -10
