"""Distill throughput benchmark on one node.

Configs (reference README.md:84-85):
  --shared : teacher + student on the same GPU(s)      (ref: 656 img/s)
  default  : service distill — teachers on their own process/GPU slice
             (ref EDL service distill: 1514 img/s whole node)

Single-GPU form (gpurun): both processes share cuda:0.

    python tools/bench_distill.py --steps 200 --batch_size 32
Service form (BASELINE config 4, 4 teacher + 4 student GPUs):
    python tools/bench_distill.py --teacher_gpus 0,1,2,3 --student_gpus 4,5,6,7
(teachers run as one process per GPU; students as one torchrun rank per
GPU; on a no-GPU box both sides fall back to CPU/gloo — the CPU smoke
path the tests use). Prints one JSON line with img/s (whole job).
"""
import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    # the threaded reader/teacher pipeline ramps for O(100) steps (queues
    # filling); short windows under-measure ~2x — window sweep in
    # profiles/r2_distill_window_sweep.json (120/250/500 steps =
    # 1156/1382/1445 img/s on one MI355X)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=30)
    ap.add_argument("--batch_size", type=int, default=32)
    ap.add_argument("--teacher_batch_size", type=int, default=16)
    ap.add_argument("--require_num", type=int, default=1)
    ap.add_argument("--teacher_model", default="resnext101_32x16d_wsl")
    ap.add_argument("--student_model", default="resnet50_vd")
    ap.add_argument("--teacher_gpus", default=None,
                    help="comma list: run the SERVICE config — one "
                         "teacher process per listed GPU")
    ap.add_argument("--student_gpus", default=None,
                    help="comma list: torchrun one student rank per GPU")
    args = ap.parse_args()
    if args.teacher_gpus is not None or args.student_gpus is not None:
        return run_service(args)

    import torch

    from edl_amd.distill.reader import DistillReader
    from edl_amd.distill.teacher_server import TeacherServer, TeacherService
    from edl_amd.train.engine import TrainerEngine

    svc = TeacherService(args.teacher_model, 1000)
    srv = TeacherServer(svc, "127.0.0.1", 0).start()
    engine = TrainerEngine(model="resnet50_vd", per_device_batch=args.batch_size,
                           base_lr=0.01, use_hip_ops=torch.cuda.is_available(),
                           graph_capture=False, kd_alpha=1.0).setup()
    engine.model.train()
    total = args.warmup + args.steps
    rng = np.random.RandomState(0)
    batches = [(rng.randn(args.batch_size, 3, 224, 224).astype(np.float32),
                rng.randint(0, 1000, (args.batch_size,)).astype(np.int64))
               for _ in range(4)]

    def batch_gen():
        for i in range(total):
            yield batches[i % len(batches)]

    dr = DistillReader(["img", "label"], ["logits"],
                       teacher_batch_size=args.teacher_batch_size,
                       require_num=args.require_num)
    dr.set_batch_generator(batch_gen)
    dr.set_fixed_teacher(["127.0.0.1:%d" % srv.port])

    n = 0
    t0 = None
    for img, label, logits in dr():
        x = torch.from_numpy(img).to(engine.device)
        if engine.device.type == "cuda":
            x = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        y = torch.from_numpy(label).to(engine.device)
        t = torch.from_numpy(np.ascontiguousarray(logits)).to(engine.device)
        engine.train_step(x, y, teacher_logits=t)
        n += 1
        if n == args.warmup:
            if engine.device.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.monotonic()
    if engine.device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    timed = n - args.warmup
    print(json.dumps({
        "metric": "img/s", "value": round(timed * args.batch_size / dt, 1),
        "unit": "img/s", "mode": "distill_shared_1gpu",
        "steps": timed, "ms_per_step": round(dt / timed * 1e3, 2),
        "teacher": args.teacher_model, "dtype": "bf16", "data": "synthetic",
        "vs_baseline_shared_656": round(timed * args.batch_size / dt / 656.0, 3),
    }))
    srv.stop()


def run_service(args):
    """BASELINE config 4: teachers on their own GPUs (one server process
    each), students data-parallel on the rest (reference EDL service
    distill, README.md:84-85: 1514 img/s on 4+4 V100)."""
    import shlex
    import subprocess

    import torch

    from edl_amd.utils.net import find_free_port

    tg = [g for g in (args.teacher_gpus or "").split(",") if g != ""]
    sg = [g for g in (args.student_gpus or "").split(",") if g != ""]
    cuda = torch.cuda.is_available()
    if not tg:
        tg = [""]  # one CPU teacher (smoke form)
    if not sg:
        sg = ["", ""]
    ports = find_free_port(len(tg))
    if len(tg) == 1:
        ports = [ports]
    procs = []
    env0 = dict(os.environ, PYTHONPATH=REPO + os.pathsep +
                os.environ.get("PYTHONPATH", ""))
    for g, port in zip(tg, ports):
        env = dict(env0, CUDA_VISIBLE_DEVICES=str(g) if cuda else "")
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "edl_amd.distill.teacher_server",
             "--model", args.teacher_model, "--host", "127.0.0.1",
             "--port", str(port)], env=env))
    teachers = ",".join("127.0.0.1:%d" % p for p in ports)
    # wait for teachers to accept
    from edl_amd.distill.registry import is_server_alive

    deadline = time.monotonic() + 240
    while time.monotonic() < deadline:
        if all(is_server_alive("127.0.0.1:%d" % p) for p in ports):
            break
        time.sleep(1.0)
    else:
        raise RuntimeError("teachers did not come up")

    env = dict(env0, CUDA_VISIBLE_DEVICES=",".join(sg) if cuda else "")
    if not cuda:
        env["EDL_FORCE_BACKEND"] = "gloo"
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", str(len(sg)), "--master-addr", "127.0.0.1",
           "--master-port", str(find_free_port()),
           os.path.join(REPO, "tools", "bench_distill_worker.py"),
           "--steps", str(args.steps), "--warmup", str(args.warmup),
           "--batch_size", str(args.batch_size),
           "--teacher_batch_size", str(args.teacher_batch_size),
           "--require_num", str(args.require_num),
           "--student_model", args.student_model,
           "--teachers", teachers]
    try:
        rc = subprocess.run(cmd, env=env).returncode
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
    return rc


if __name__ == "__main__":
    sys.exit(main() or 0)
