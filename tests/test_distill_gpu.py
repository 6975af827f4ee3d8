"""GPU end-to-end distill: ResNeXt teacher server + ResNet50_vd student
with the fused KD loss, teacher and student sharing cuda:0 (the
reference's same-GPU distill config, README.md:84)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def test_distill_end_to_end_one_gpu():
    from edl_amd.distill.reader import DistillReader
    from edl_amd.distill.teacher_server import TeacherServer, TeacherService
    from edl_amd.train.engine import TrainerEngine

    svc = TeacherService("resnext101_32x16d_wsl", num_classes=1000)
    srv = TeacherServer(svc, host="127.0.0.1", port=0).start()
    try:
        engine = TrainerEngine(model="resnet50_vd", per_device_batch=8,
                               base_lr=0.01, use_hip_ops=True,
                               graph_capture=False, kd_alpha=1.0).setup()
        engine.model.train()

        def batch_gen():
            rng = np.random.RandomState(0)
            for _ in range(4):
                yield (rng.randn(8, 3, 224, 224).astype(np.float32),
                       rng.randint(0, 1000, (8,)).astype(np.int64))

        dr = DistillReader(ins=["img", "label"], predicts=["logits"],
                           teacher_batch_size=8, require_num=1)
        dr.set_batch_generator(batch_gen)
        dr.set_fixed_teacher(["127.0.0.1:%d" % srv.port])
        n = 0
        for img, label, logits in dr():
            x = torch.from_numpy(img).cuda().to(torch.bfloat16).contiguous(
                memory_format=torch.channels_last)
            y = torch.from_numpy(label).cuda()
            t = torch.from_numpy(np.ascontiguousarray(logits)).cuda()
            loss = engine.train_step(x, y, teacher_logits=t)
            assert torch.isfinite(loss), loss
            n += 1
        torch.cuda.synchronize()
        assert n == 4
    finally:
        srv.stop()
