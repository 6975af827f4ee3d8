#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c5
echo "== s2 dgrad numerics + conv suite =="
timeout 900 python -m pytest tests/test_gemm_gpu.py -q -m gpu > gpurun_out/r2c5/pytest_gemm.log 2>&1
echo "gemm pytest rc=$?"
echo "== bench.py (s2 dgrad in the student step) =="
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c5/bench1.log 2>&1
echo "bench rc=$?"
echo "== distill shared (ref 656) =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c5/distill_shared.log 2>&1
echo "distill rc=$?"
echo "== rocprof stats (csv output this time) =="
mkdir -p gpurun_out/r2c5/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c5/prof -o r2s -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 15 ) > gpurun_out/r2c5/rocprof.log 2>&1
echo "rocprof rc=$?"
ls gpurun_out/r2c5/prof/ | head
echo "== tails =="
for f in gpurun_out/r2c5/pytest_gemm.log gpurun_out/r2c5/bench1.log gpurun_out/r2c5/distill_shared.log; do echo "--- $f"; tail -4 "$f"; done
head -40 gpurun_out/r2c5/prof/*kernel_stats* 2>/dev/null
