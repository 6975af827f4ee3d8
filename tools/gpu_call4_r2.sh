#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c4
echo "== bnrelu GPU numerics (v2 finalize paths) =="
timeout 600 python -m pytest tests/test_bnrelu.py tests/test_ops_gpu.py -q -m gpu > gpurun_out/r2c4/pytest_bn_ops.log 2>&1
echo "bn/ops pytest rc=$?"
echo "== distill shared-GPU bench (headline: ref 656 img/s) =="
timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c4/distill_shared.log 2>&1
echo "distill rc=$?"
echo "== bench.py end-to-end (BN v2 + fixes) =="
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 15 > gpurun_out/r2c4/bench1.log 2>&1
echo "bench rc=$?"
echo "== rocprof step trace for profiles/ =="
cd /tmp && export TMPDIR=/tmp && cd - > /dev/null
timeout 900 bash -c 'cd /tmp && rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r2c4/prof -o r2step -- python /root/repo/bench.py --gpus 1 --steps 20 --warmup 15' > gpurun_out/r2c4/rocprof.log 2>&1
echo "rocprof rc=$?"
ls gpurun_out/r2c4/prof* 2>/dev/null | head
echo "== tails =="
for f in gpurun_out/r2c4/pytest_bn_ops.log gpurun_out/r2c4/distill_shared.log gpurun_out/r2c4/bench1.log gpurun_out/r2c4/rocprof.log; do echo "--- $f"; tail -6 "$f"; done
