#!/bin/bash
# r2 call 29: validate zero_src restriction; stats tests + bench + finalize timing.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c29
timeout 600 python -m pytest tests/test_bnrelu.py tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c29/pytest.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c29/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c29/bench2.json
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c29/prof -o zs -- python bench.py --gpus 1 --steps 30 --warmup 6 > gpurun_out/r2c29/bench_prof.log 2>&1
python - <<'PY'
import csv, glob
f = glob.glob('gpurun_out/r2c29/prof/*kernel_stats.csv')
for row in csv.DictReader(open(f[0])):
    n = row.get('Name','')
    if 'bn_finalize' in n or 'FillFunctor' in n or 'cast_bf16' in n:
        print(n[:60], row['Calls'], round(int(row['TotalDurationNs'])/1e3/36,1), "us/step")
PY
cat gpurun_out/r2c29/pytest.txt gpurun_out/r2c29/bench1.json gpurun_out/r2c29/bench2.json
