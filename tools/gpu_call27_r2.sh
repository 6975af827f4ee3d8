#!/bin/bash
# r2 call 27: validate split-K pool + full suite + bench + fill count.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c27
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c27/pytest.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c27/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c27/bench2.json
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c27/prof -o sk -- python bench.py --gpus 1 --steps 30 --warmup 6 > gpurun_out/r2c27/bench_prof.log 2>&1
tail -1 gpurun_out/r2c27/bench_prof.log
python - <<'PY'
import csv, glob
f = glob.glob('gpurun_out/r2c27/prof/*kernel_stats.csv')
fills = fillns = tot = 0
cast = castns = 0
for row in csv.DictReader(open(f[0])):
    ns = int(row.get('TotalDurationNs', 0)); tot += ns
    if 'FillFunctor' in row.get('Name',''):
        fills += int(row['Calls']); fillns += ns
    if 'cast_bf16_zero' in row.get('Name',''):
        cast += int(row['Calls']); castns += ns
print("fills", fills, "fill_us", fillns/1e3, "cast_calls", cast,
      "cast_us", castns/1e3, "total_ms", tot/1e6)
PY
cat gpurun_out/r2c27/pytest.txt gpurun_out/r2c27/bench1.json gpurun_out/r2c27/bench2.json
