#!/bin/bash
# Round-2 GPU call 1: multi-rank RCCL de-risk on one MI355X.
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/derisk
echo "== smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/derisk/smoke.log 2>&1
echo "smoke rc=$?"
echo "== multirank pytest (2 RCCL ranks on cuda:0 + graph-LR) =="
timeout 600 python -m pytest tests/test_multirank_gpu.py -x -q -m gpu > gpurun_out/derisk/pytest_multirank.log 2>&1
echo "pytest rc=$?"
echo "== bench world=2 RCCL (graph off default) =="
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29533 bench.py --gpus 2 --steps 10 --warmup 8 \
  > gpurun_out/derisk/bench_w2.log 2>&1
echo "bench_w2 rc=$?"
echo "== bench world=2 RCCL inside hipGraph (A/B) =="
EDL_GRAPH_CAPTURE=1 timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29535 bench.py --gpus 2 --steps 10 --warmup 8 \
  > gpurun_out/derisk/bench_w2_graph.log 2>&1
echo "bench_w2_graph rc=$?"
echo "== tails =="
for f in gpurun_out/derisk/*.log; do echo "--- $f"; tail -4 "$f"; done
