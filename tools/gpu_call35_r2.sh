#!/bin/bash
# r2 call 35: validate pow2 BN addressing + bench; elastic refresh.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c35
timeout 600 python -m pytest tests/test_bnrelu.py tests/test_ops_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c35/pytest.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c35/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c35/bench2.json
timeout 600 python tools/bench_elastic.py --share-gpu0 --steps 12 2>&1 | tail -4 > gpurun_out/r2c35/elastic.txt
cat gpurun_out/r2c35/pytest.txt gpurun_out/r2c35/bench1.json gpurun_out/r2c35/bench2.json gpurun_out/r2c35/elastic.txt
