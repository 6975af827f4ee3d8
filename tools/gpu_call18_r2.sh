#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c18
echo "== numerics =="
timeout 900 python -m pytest tests/test_bnrelu.py tests/test_ops_gpu.py -q -m gpu > gpurun_out/r2c18/pytest.log 2>&1
echo "pytest rc=$?"
echo "== bench A/B x2 each =="
for i in 1 2; do
  timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c18/bench_fused_$i.log 2>&1
  EDL_BN_STATS_FUSED=0 timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c18/bench_unfused_$i.log 2>&1
done
echo benches done
for f in gpurun_out/r2c18/*.log; do echo "--- $f"; tail -2 "$f" | grep -oE '"value": [0-9.]+|passed|failed' | head -3; done
