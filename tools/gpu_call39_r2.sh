#!/bin/bash
# r2 call 39: comprehensive final evidence on the end-of-round tree.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c39
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c39/pytest.txt
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" 2>&1 | tail -2 > gpurun_out/r2c39/smoke.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c39/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c39/bench2.json
# ablation: all torch/library paths (same recipe as profiles/README r01_step7)
EDL_FUSED_BN=0 EDL_CONV1X1=matmul EDL_CONV3X3=miopen EDL_CONV3X3_SMALL=0 \
  timeout 420 python bench.py --gpus 1 --steps 40 --warmup 12 --use_hip_ops 0 2>&1 | tail -1 > gpurun_out/r2c39/bench_library.json
timeout 300 python tools/bench_ctr.py --steps 200 --warmup 30 2>&1 | tail -1 > gpurun_out/r2c39/ctr.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 --dtype fp16 2>&1 | tail -1 > gpurun_out/r2c39/bench_fp16.json
timeout 300 python tools/teacher_fwd_bench.py --iters 30 --warmup 8 2>&1 | tail -2 > gpurun_out/r2c39/teacher_fwd.txt
cat gpurun_out/r2c39/pytest.txt gpurun_out/r2c39/smoke.txt gpurun_out/r2c39/bench1.json gpurun_out/r2c39/bench2.json gpurun_out/r2c39/bench_library.json gpurun_out/r2c39/ctr.json gpurun_out/r2c39/bench_fp16.json gpurun_out/r2c39/teacher_fwd.txt
