#!/bin/bash
# r2 call 28: distill evidence on the latest tree (4 runs one box).
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c28
for i in 1 2 3 4; do
  timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c28/distill_$i.log 2>&1
done
for f in gpurun_out/r2c28/distill_*.log; do echo "--- $f"; tail -2 "$f" | grep -v amdgpu; done
