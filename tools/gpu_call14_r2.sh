#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c14
echo "== re-validate maxpool/dgc tests =="
timeout 600 python -m pytest tests/test_ops_gpu.py -q -m gpu -k maxpool > gpurun_out/r2c14/pytest_mp.log 2>&1
echo "mp rc=$?"
timeout 600 python -m pytest tests/test_gpu_extras.py::test_dgc_two_rank_cuda -q -m gpu > gpurun_out/r2c14/pytest_dgc.log 2>&1
echo "dgc rc=$?"
echo "== tn probe timings =="
timeout 600 python tools/tn_pmc_probe.py > gpurun_out/r2c14/tn_times.log 2>&1
echo "tn rc=$?"
echo "== tn PMC =="
mkdir -p gpurun_out/r2c14/pmc
( cd /tmp && export TMPDIR=/tmp && timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_LDS_BANK_CONFLICT,SQ_LDS_IDX_ACTIVE,FETCH_SIZE --kernel-trace --output-format csv -d /root/repo/gpurun_out/r2c14/pmc -o tnpmc -- python /root/repo/tools/tn_pmc_probe.py ) > gpurun_out/r2c14/pmc.log 2>&1
echo "pmc rc=$?"
ls gpurun_out/r2c14/pmc/
for f in gpurun_out/r2c14/pytest_mp.log gpurun_out/r2c14/pytest_dgc.log gpurun_out/r2c14/tn_times.log; do echo "--- $f"; tail -6 "$f" | grep -v amdgpu; done
