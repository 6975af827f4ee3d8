#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c6
echo "== bn bwd sweep (contig variants) =="
timeout 900 python tools/bn_bwd_bench.py --iters 20 > gpurun_out/r2c6/bn_bwd_bench.log 2>&1
echo "bnbwd rc=$?"
echo "== gpu extras (fp16/ctr/dgc) =="
timeout 900 python -m pytest tests/test_gpu_extras.py tests/test_bnrelu.py -q -m gpu > gpurun_out/r2c6/pytest_extras.log 2>&1
echo "extras rc=$?"
echo "== bench contig A/B =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c6/bench_base.log 2>&1
echo "base rc=$?"
EDL_BN_CONTIG=1 timeout 420 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c6/bench_contig.log 2>&1
echo "contig rc=$?"
echo "== fp16 bench point =="
timeout 420 python bench.py --gpus 1 --steps 20 --warmup 15 --dtype fp16 > gpurun_out/r2c6/bench_fp16.log 2>&1
echo "fp16 rc=$?"
echo "== tails =="
for f in gpurun_out/r2c6/*.log; do echo "--- $f"; tail -4 "$f" | grep -vE "amdgpu.ids"; done
grep -h '"bench": "bn_bwd"' gpurun_out/r2c6/bn_bwd_bench.log
