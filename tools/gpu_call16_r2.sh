#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c16
echo "== FULL gpu suite =="
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/r2c16/pytest_all.log 2>&1
echo "full rc=$?"
echo "== smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r2c16/smoke.log 2>&1
echo "smoke rc=$?"
echo "== bench x2 =="
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c16/bench1.log 2>&1
echo "b1 rc=$?"
timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c16/bench2.log 2>&1
echo "b2 rc=$?"
echo "== distill x2 =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c16/distill1.log 2>&1
echo "d1 rc=$?"
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c16/distill2.log 2>&1
echo "d2 rc=$?"
echo "== final step trace =="
mkdir -p gpurun_out/r2c16/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c16/prof -o r2final -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 10 ) > gpurun_out/r2c16/rocprof.log 2>&1
echo "prof rc=$?"
echo "== wgrad microbench for profiles =="
timeout 420 python tools/gemm_bench.py --wgrad > gpurun_out/r2c16/wgrad_bench.log 2>&1
echo "wg rc=$?"
for f in gpurun_out/r2c16/pytest_all.log gpurun_out/r2c16/smoke.log gpurun_out/r2c16/bench1.log gpurun_out/r2c16/bench2.log gpurun_out/r2c16/distill1.log gpurun_out/r2c16/distill2.log; do echo "--- $f"; tail -2 "$f" | grep -v amdgpu; done
