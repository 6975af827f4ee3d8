#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c3
echo "== teacher fwd A/B sweep (grouped conv routing) =="
timeout 700 python tools/teacher_fwd_bench.py --batch 16 --iters 10 > gpurun_out/r2c3/teacher_fwd_b16.log 2>&1
echo "teacher16 rc=$?"
timeout 700 python tools/teacher_fwd_bench.py --batch 32 --iters 10 > gpurun_out/r2c3/teacher_fwd_b32.log 2>&1
echo "teacher32 rc=$?"
echo "== bn bwd microbench sweep =="
timeout 900 python tools/bn_bwd_bench.py --iters 20 > gpurun_out/r2c3/bn_bwd_bench.log 2>&1
echo "bnbwd rc=$?"
echo "== bn numerics quick =="
timeout 600 python -m pytest tests/test_ops_gpu.py -q -m gpu -k "bn" > gpurun_out/r2c3/pytest_bn.log 2>&1
echo "bn pytest rc=$?"
echo "== tails =="
for f in gpurun_out/r2c3/*.log; do echo "--- $f"; tail -30 "$f"; done
