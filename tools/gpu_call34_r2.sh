#!/bin/bash
# r2 call 34: sustained-training stability soak + re-verify G3S clamp fix.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c34
timeout 300 python -m pytest tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -2 > gpurun_out/r2c34/pytest.txt
# 4-minute sustained run with checkpointing: throughput + memory stability
python - <<'PY' > gpurun_out/r2c34/soak.txt 2>&1
import time, torch, os, tempfile
from edl_amd.data.synthetic import SyntheticImageNet
from edl_amd.train.engine import TrainerEngine

ckdir = tempfile.mkdtemp(prefix="soak_ck_")
eng = TrainerEngine(model="resnet50_vd", per_device_batch=32, dtype="bf16",
                    channels_last=True, checkpoint_dir=ckdir).setup()
eng.model.train()
loader = SyntheticImageNet(32, eng.device, channels_last=True, seed=3)
x, y = loader.next()
eng.maybe_capture(x, y)
for _ in range(8):
    x, y = loader.next(); eng.replay_step(x, y)
torch.cuda.synchronize()
t_end = time.monotonic() + 240
steps = 0; window = time.monotonic(); wsteps = 0
while time.monotonic() < t_end:
    x, y = loader.next()
    eng.replay_step(x, y)
    steps += 1; wsteps += 1
    if steps % 1000 == 0:
        torch.cuda.synchronize()
        now = time.monotonic()
        print(f"step {steps}: {wsteps*32/(now-window):.0f} img/s, "
              f"mem {torch.cuda.max_memory_allocated()/2**30:.2f} GiB",
              flush=True)
        window = now; wsteps = 0
    if steps % 5000 == 0:
        eng.save_checkpoint(epoch=0, extra={"step": steps})
torch.cuda.synchronize()
print(f"SOAK OK: {steps} steps, peak mem "
      f"{torch.cuda.max_memory_allocated()/2**30:.2f} GiB, "
      f"checkpoints at {sorted(os.listdir(ckdir))[-2:]}")
PY
tail -12 gpurun_out/r2c34/soak.txt
cat gpurun_out/r2c34/pytest.txt
