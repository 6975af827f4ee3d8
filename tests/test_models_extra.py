"""MNIST + NLP distill model families (reference
example/distill/mnist_distill/train_with_fleet.py nn_type nets and
example/distill/nlp/model.py BOW/CNN students)."""
import pytest
import torch

from edl_amd.models import build_model
from edl_amd.models.text import TextBOW, TextCNN
from edl_amd.ops.functional import kd_soft_cross_entropy


@pytest.mark.parametrize("name", ["mnist_cnn", "mnist_mlp", "mnist_softmax"])
def test_mnist_nets_forward_backward(name):
    torch.manual_seed(0)
    m = build_model(name)
    x = torch.randn(4, 1, 28, 28)
    y = m(x)
    assert y.shape == (4, 10)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_text_bow_pad_invariant():
    """BOW: appending pad tokens (id 0) must not change the logits — the
    pad mask + padding_idx-zero embeddings define the reference semantics
    (nlp/model.py pad_mask reduce_sum). The CNN is NOT strictly invariant
    (its conv bias leaks into boundary pooling windows — true of the
    reference fluid CNN as well), so it gets shape/backward coverage."""
    torch.manual_seed(1)
    m = TextBOW(vocab_size=50, num_classes=2)
    ids = torch.randint(1, 50, (3, 7))
    padded = torch.cat([ids, torch.zeros(3, 5, dtype=torch.long)], dim=1)
    y1 = m(ids)
    y2 = m(padded)
    assert y1.shape == (3, 2)
    assert torch.allclose(y1, y2, atol=1e-5), (y1 - y2).abs().max().item()


def test_text_cnn_forward_backward():
    torch.manual_seed(1)
    m = TextCNN(vocab_size=50, num_classes=2)
    ids = torch.randint(1, 50, (3, 7))
    y = m(ids)
    assert y.shape == (3, 2)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters()
               if p.requires_grad)


def test_text_kd_student_learns_teacher():
    """CNN student distilled toward a frozen BOW teacher's soft labels via
    the same KD soft-label CE the resnet/mnist examples use: the KD loss
    must drop substantially."""
    torch.manual_seed(2)
    teacher = TextBOW(vocab_size=64, num_classes=2)
    for p in teacher.parameters():
        p.requires_grad_(False)
    student = TextCNN(vocab_size=64, num_classes=2)
    opt = torch.optim.Adam(student.parameters(), lr=5e-3)
    ids = torch.randint(1, 64, (64, 9))
    with torch.no_grad():
        # sharpen: a random-init teacher is near-uniform (KD floor ln 2)
        t_logits = teacher(ids) * 8.0
    losses = []
    for _ in range(150):
        opt.zero_grad()
        loss = kd_soft_cross_entropy(student(ids), t_logits)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.6, (losses[0], losses[-1])
    # student's hard predictions match the teacher's on the training set
    agree = (student(ids).argmax(1) == t_logits.argmax(1)).float().mean()
    assert agree > 0.9, float(agree)


def test_text_distill_pipeline_real_teacher():
    """NLP distill end-to-end on CPU: a served TextBOW teacher feeds soft
    labels through the full DistillReader pipeline (reference
    example/distill/nlp/distill.py flow) and a token-CNN student trains
    one pass against them."""
    import numpy as np

    from edl_amd.distill.reader import DistillReader
    from edl_amd.distill.teacher_server import TeacherServer, TeacherService

    torch.manual_seed(3)
    teacher = TextBOW(vocab_size=32, num_classes=2)
    srv = TeacherServer(TeacherService(model=teacher,
                                       device=torch.device("cpu")),
                        host="127.0.0.1", port=0).start()
    try:
        rng = np.random.RandomState(0)
        n = 24

        def gen():
            for i in range(n):
                yield (rng.randint(1, 32, size=(6,)).astype(np.int64),
                       np.int64(i % 2))

        dr = DistillReader(ins=["ids", "y"], predicts=["logits"],
                           teacher_batch_size=4, require_num=1)
        dr.set_sample_generator(gen)
        dr.set_fixed_teacher(["127.0.0.1:%d" % srv.port])
        student = TextCNN(vocab_size=32, num_classes=2)
        opt = torch.optim.SGD(student.parameters(), lr=0.1)
        seen = 0
        batch = []
        for ids, y, logits in dr():
            ref = teacher(torch.from_numpy(np.ascontiguousarray(ids))
                          .unsqueeze(0)).detach().numpy()[0]
            assert np.allclose(logits, ref, atol=1e-4)  # served == local
            batch.append((ids, logits))
            seen += 1
            if len(batch) == 8:
                x = torch.from_numpy(np.stack([b[0] for b in batch]))
                t = torch.from_numpy(np.stack([b[1] for b in batch]))
                opt.zero_grad()
                kd_soft_cross_entropy(student(x), t * 4.0).backward()
                opt.step()
                batch = []
        assert seen == n
    finally:
        srv.stop()
