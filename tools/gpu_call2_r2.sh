#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/derisk gpurun_out/grouped
echo "== grouped conv numerics (all widths) =="
timeout 600 python -m pytest "tests/test_gemm_gpu.py::test_conv3x3_grouped_eval_numerics" -q -m gpu > gpurun_out/grouped/pytest_grouped.log 2>&1
echo "grouped pytest rc=$?"
echo "== teacher fwd A/B sweep =="
timeout 600 python tools/teacher_fwd_bench.py --batch 16 --iters 10 > gpurun_out/grouped/teacher_fwd.log 2>&1
echo "teacher rc=$?"
timeout 600 python tools/teacher_fwd_bench.py --batch 32 --iters 10 > gpurun_out/grouped/teacher_fwd_b32.log 2>&1
echo "teacher32 rc=$?"
echo "== multirank derisk pytest =="
timeout 900 python -m pytest tests/test_multirank_gpu.py -x -q -m gpu > gpurun_out/derisk/pytest_multirank.log 2>&1
echo "pytest rc=$?"
echo "== bench world=2 on one GPU (gloo collectives) =="
EDL_FORCE_BACKEND=gloo timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29533 bench.py --gpus 2 --steps 10 --warmup 8 \
  > gpurun_out/derisk/bench_w2_gloo.log 2>&1
echo "bench_w2_gloo rc=$?"
echo "== tails =="
for f in gpurun_out/grouped/*.log gpurun_out/derisk/pytest_multirank.log gpurun_out/derisk/bench_w2_gloo.log; do echo "--- $f"; tail -8 "$f"; done
