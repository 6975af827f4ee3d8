#!/bin/bash
# r2 call 31: comprehensive validation of the final-ish tree.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c31
timeout 900 python -m pytest tests -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c31/pytest.txt
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" 2>&1 | tail -2 > gpurun_out/r2c31/smoke.txt
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c31/bench1.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c31/bench2.json
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c31/prof -o fin -- python bench.py --gpus 1 --steps 30 --warmup 6 > gpurun_out/r2c31/bench_prof.log 2>&1
tail -1 gpurun_out/r2c31/bench_prof.log
cat gpurun_out/r2c31/pytest.txt gpurun_out/r2c31/smoke.txt gpurun_out/r2c31/bench1.json gpurun_out/r2c31/bench2.json
