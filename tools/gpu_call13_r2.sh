#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c13
echo "== FULL gpu suite =="
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/r2c13/pytest_all.log 2>&1
echo "full pytest rc=$?"
echo "== elastic resize-recovery ON GPU (2->1->2, gloo shared cuda:0) =="
timeout 600 python tools/bench_elastic.py --start 2 --drop-to 1 --rejoin --share-gpu0 > gpurun_out/r2c13/elastic_gpu.log 2>&1
echo "elastic rc=$?"
echo "== service distill on one GPU (teacher proc + student rank, config-4 orchestration) =="
timeout 900 python tools/bench_distill.py --steps 15 --warmup 5 --teacher_gpus 0 --student_gpus 0 > gpurun_out/r2c13/distill_service.log 2>&1
echo "service rc=$?"
for f in gpurun_out/r2c13/*.log; do echo "--- $f"; tail -4 "$f" | grep -v amdgpu.ids; done
