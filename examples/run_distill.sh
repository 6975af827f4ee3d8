#!/bin/bash
# BASELINE config 4: ResNeXt101_32x16d_wsl teachers serving ResNet50_vd
# students. Run teacher(s) on their GPUs, a discovery server, then students.
set -e
# run from anywhere: the repo root is importable
export PYTHONPATH="$(cd "$(dirname "$0")/.." && pwd)${PYTHONPATH:+:$PYTHONPATH}"
STORE=${STORE:-127.0.0.1:2379}
python -m edl_amd.coord.server --port "${STORE##*:}" &
sleep 1
for GPU in 0 1 2 3; do
  CUDA_VISIBLE_DEVICES=$GPU python -m edl_amd.distill.teacher_server \
      --port $((9292 + GPU)) --service_name resnext_teacher \
      --store_endpoints "$STORE" &
done
STORE="$STORE" python - <<'PY' &
import os
import threading

from edl_amd.distill.discovery import DiscoveryServer

DiscoveryServer(os.environ["STORE"]).start()
threading.Event().wait()
PY
CUDA_VISIBLE_DEVICES=4,5,6,7 python -m edl_amd.launch \
    --store_endpoints "$STORE" --nodes_range 1:1 --job_id distill \
    -- -m edl_amd.train.train_distill --balance_server "$STORE" \
       --service_name resnext_teacher --require_num 2
