#!/bin/bash
# r2 call 32: PMC counters for the final-state hot kernels + prefill test.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c32/pmcA gpurun_out/r2c32/pmcB
timeout 300 python -m pytest tests/test_gpu_extras.py -m gpu -q -p no:cacheprovider 2>&1 | tail -2 > gpurun_out/r2c32/pytest.txt
( cd /tmp && export TMPDIR=/tmp && timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_INSTS_MFMA,FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2c32/pmcA -o a -- python /root/repo/bench.py --gpus 1 --steps 6 --warmup 2 ) > gpurun_out/r2c32/pmcA.log 2>&1
echo "pmcA rc=$?"
( cd /tmp && export TMPDIR=/tmp && timeout 600 rocprofv3 --pmc WRITE_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2c32/pmcB -o b -- python /root/repo/bench.py --gpus 1 --steps 6 --warmup 2 ) > gpurun_out/r2c32/pmcB.log 2>&1
echo "pmcB rc=$?"
python - <<'PY' > gpurun_out/r2c32/pmc_summary.txt 2>&1
import csv, glob, collections
def load(d):
    per = collections.defaultdict(lambda: collections.defaultdict(float))
    calls = collections.Counter()
    for f in glob.glob(d + '/*counter_collection.csv'):
        for row in csv.DictReader(open(f)):
            k = row.get('Kernel_Name', '')[:56]
            per[k][row['Counter_Name']] += float(row['Counter_Value'])
            calls[(k, row['Counter_Name'])] += 1
    return per
A = load('gpurun_out/r2c32/pmcA'); B = load('gpurun_out/r2c32/pmcB')
# durations from kernel trace
dur = collections.Counter()
for f in glob.glob('gpurun_out/r2c32/pmcA/*kernel_trace.csv'):
    for row in csv.DictReader(open(f)):
        k = row.get('Kernel_Name', '')[:56]
        dur[k] += int(row['End_Timestamp']) - int(row['Start_Timestamp'])
print(f"{'kernel':56s} {'ms_tot':>7s} {'mfma/cyc':>8s} {'waitA%':>6s} {'waitI%':>6s} {'HBM GB/s':>8s}")
for k, ns in dur.most_common(14):
    a = A.get(k, {}); b = B.get(k, {})
    wc = a.get('SQ_WAVE_CYCLES', 0) or 1
    hbm = (a.get('FETCH_SIZE', 0) + b.get('WRITE_SIZE', 0)) * 1024
    print(f"{k:56s} {ns/1e6:7.2f} {a.get('SQ_INSTS_MFMA',0)/wc:8.4f} "
          f"{100*a.get('SQ_WAIT_ANY',0)/wc:6.1f} {100*a.get('SQ_WAIT_INST_ANY',0)/wc:6.1f} "
          f"{hbm/max(ns,1):8.1f}")
PY
cat gpurun_out/r2c32/pytest.txt gpurun_out/r2c32/pmc_summary.txt
