#!/bin/bash
# r2 call 23: validate pooled partials + zero-after-read (numerics x2 steps,
# capped shapes), then bench x2.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c23
timeout 600 python -m pytest tests/test_bnrelu.py tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c23/pytest.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c23/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c23/bench2.json
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c23/prof -o pool -- python bench.py --gpus 1 --steps 30 --warmup 6 > gpurun_out/r2c23/bench_prof.log 2>&1
tail -1 gpurun_out/r2c23/bench_prof.log
# count remaining elementwise fills in the trace
python - <<'PY'
import csv, glob
f = glob.glob('gpurun_out/r2c23/prof/*kernel_stats.csv')
if f:
    for row in csv.DictReader(open(f[0])):
        nm = row.get('Name', '')
        if 'FillFunctor' in nm or 'elementwise' in nm or 'bn_finalize' in nm:
            print(row.get('Name','')[:70], row.get('Calls'), row.get('TotalDurationNs'))
PY
cat gpurun_out/r2c23/pytest.txt gpurun_out/r2c23/bench1.json gpurun_out/r2c23/bench2.json
