#!/bin/bash
# Final round-2 validation sweep.
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c21
echo "== FULL gpu suite x2 (flake check) =="
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/r2c21/pytest1.log 2>&1
echo "p1 rc=$?"
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/r2c21/pytest2.log 2>&1
echo "p2 rc=$?"
echo "== smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/r2c21/smoke.log 2>&1
echo "smoke rc=$?"
echo "== mid-epoch kill/resume =="
CKPT=/tmp/soak2; rm -rf $CKPT
timeout --signal=KILL 35 python -m edl_amd.train.train_resnet \
  --model resnet50_vd --batch_size 32 --num_epochs 1 --steps_per_epoch 20000 \
  --checkpoint $CKPT --checkpoint_steps 200 > gpurun_out/r2c21/soak1.log 2>&1
echo "killed rc=$? (137 expected)"
timeout --signal=KILL 25 python -m edl_amd.train.train_resnet \
  --model resnet50_vd --batch_size 32 --num_epochs 1 --steps_per_epoch 20000 \
  --checkpoint $CKPT --checkpoint_steps 200 > gpurun_out/r2c21/soak2.log 2>&1
echo "resume rc=$? (137 expected)"
grep -E "resumed|step " gpurun_out/r2c21/soak2.log | head -3 > gpurun_out/r2c21/resume_evidence.log
grep -E "step |img/s" gpurun_out/r2c21/soak1.log | tail -2 >> gpurun_out/r2c21/resume_evidence.log
echo "== bench x3 =="
for i in 1 2 3; do
  timeout 420 python bench.py --gpus 1 --steps 30 --warmup 10 > gpurun_out/r2c21/bench_$i.log 2>&1
done
echo "== distill x3 =="
for i in 1 2 3; do
  timeout 900 python tools/bench_distill.py --steps 20 --warmup 5 > gpurun_out/r2c21/distill_$i.log 2>&1
done
echo "== final trace =="
mkdir -p gpurun_out/r2c21/prof
( cd /tmp && export TMPDIR=/tmp && timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/r2c21/prof -o r2end -- python /root/repo/bench.py --gpus 1 --steps 22 --warmup 10 ) > gpurun_out/r2c21/rocprof.log 2>&1
echo "trace rc=$?"
for f in gpurun_out/r2c21/pytest1.log gpurun_out/r2c21/pytest2.log gpurun_out/r2c21/smoke.log gpurun_out/r2c21/resume_evidence.log gpurun_out/r2c21/bench_*.log gpurun_out/r2c21/distill_*.log; do echo "--- $f"; tail -2 "$f" | grep -v amdgpu; done
