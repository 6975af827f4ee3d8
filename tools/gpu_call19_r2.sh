#!/bin/bash
# Long-run stability soak: 300 steps with step-checkpoints, hard kill
# mid-epoch, resume, verify continuation + memory stability.
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c19
CKPT=/tmp/soak_ckpt
rm -rf $CKPT
echo "== phase 1: train 300 steps, ckpt every 50, killed at ~210s =="
timeout --signal=KILL 210 python -m edl_amd.train.train_resnet \
  --model resnet50_vd --batch_size 32 --num_epochs 1 --steps_per_epoch 300 \
  --checkpoint $CKPT --checkpoint_steps 50 > gpurun_out/r2c19/phase1.log 2>&1
echo "phase1 rc=$? (137=killed as planned)"
ls $CKPT > gpurun_out/r2c19/ckpts_after_kill.log 2>&1
echo "== phase 2: resume, finish the epoch =="
timeout 420 python -m edl_amd.train.train_resnet \
  --model resnet50_vd --batch_size 32 --num_epochs 1 --steps_per_epoch 300 \
  --checkpoint $CKPT --checkpoint_steps 50 > gpurun_out/r2c19/phase2.log 2>&1
echo "phase2 rc=$?"
echo "--- phase1 tail:"; grep -E "step|resumed|img/s|mem" gpurun_out/r2c19/phase1.log | tail -5
echo "--- phase2 head:"; grep -E "resumed|fresh" gpurun_out/r2c19/phase2.log | head -3
echo "--- phase2 tail:"; tail -4 gpurun_out/r2c19/phase2.log | grep -v amdgpu
echo "--- checkpoints:"; cat gpurun_out/r2c19/ckpts_after_kill.log
