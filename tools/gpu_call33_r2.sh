#!/bin/bash
# r2 call 33: G3S stem wgrad numerics + A/B.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out/r2c33
timeout 600 python -m pytest tests/test_gemm_gpu.py -m gpu -q -p no:cacheprovider 2>&1 | tail -3 > gpurun_out/r2c33/pytest.txt
# stem wgrad micro: time the three shapes under both routes
python - <<'PY' > gpurun_out/r2c33/stem_ab.txt 2>&1
import os, torch, time
from edl_amd import ops
import torch.nn.functional as F
e = ops.ext()
shapes = [(32, 3, 32, 224, 2), (32, 32, 32, 112, 1), (32, 32, 64, 112, 1)]
for n, ci, co, hw, stride in shapes:
    x = torch.randn(n, ci, hw, hw, device="cuda").to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    ho = (hw - 1) // stride + 1
    dy = torch.randn(n, co, ho, ho, device="cuda").to(torch.bfloat16)
    dy = dy.contiguous(memory_format=torch.channels_last)
    dy2d = dy.permute(0, 2, 3, 1).reshape(-1, co)
    if co < 64: dy2d = F.pad(dy2d, (0, 64 - co))
    dy2d = dy2d.contiguous()
    def tn():
        return e.gemm_tn3x3_small(dy2d, x, stride)
    def mi():
        return torch.nn.grad.conv2d_weight(x, (co, ci, 3, 3), dy,
                                           stride=(stride, stride), padding=(1, 1))
    for name, fn in (("tn", tn), ("miopen", mi)):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.monotonic()
        for _ in range(30): fn()
        torch.cuda.synchronize()
        us = (time.monotonic() - t0) / 30 * 1e6
        print(f"({n},{ci},{co},{hw},s{stride}) {name:6s} {us:8.1f} us", flush=True)
PY
EDL_STEM_WGRAD=tn timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c33/bench_tn.json
timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c33/bench_miopen.json
EDL_STEM_WGRAD=tn timeout 420 python bench.py --gpus 1 --steps 60 --warmup 8 2>&1 | tail -1 > gpurun_out/r2c33/bench_tn2.json
cat gpurun_out/r2c33/pytest.txt gpurun_out/r2c33/stem_ab.txt gpurun_out/r2c33/bench_*.json
