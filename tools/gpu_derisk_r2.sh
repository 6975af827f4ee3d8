#!/bin/bash
# Round-2 GPU de-risk on one MI355X (revised after the
# measured RCCL Duplicate-GPU refusal — see tools/rccl_probe.py).
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/derisk
echo "== multirank pytest (world-1 RCCL + graph, refusal evidence, gloo engine path) =="
timeout 900 python -m pytest tests/test_multirank_gpu.py -x -q -m gpu > gpurun_out/derisk/pytest_multirank.log 2>&1
echo "pytest rc=$?"
echo "== bench world=2 on one GPU (gloo collectives, HIP compute) =="
EDL_FORCE_BACKEND=gloo timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29533 bench.py --gpus 2 --steps 10 --warmup 8 \
  > gpurun_out/derisk/bench_w2_gloo.log 2>&1
echo "bench_w2_gloo rc=$?"
echo "== tails =="
for f in gpurun_out/derisk/pytest_multirank.log gpurun_out/derisk/bench_w2_gloo.log; do echo "--- $f"; tail -4 "$f"; done
