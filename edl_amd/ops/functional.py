"""Autograd wrappers over the HIP kernels + torch reference fallbacks."""
import torch

from . import available, ext


class _KDSoftCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, student_logits, teacher_logits):
        ctx.save_for_backward(student_logits, teacher_logits)
        lpr = ext().kd_ce_forward(student_logits, teacher_logits)
        return lpr.mean()

    @staticmethod
    def backward(ctx, gout):
        s, t = ctx.saved_tensors
        gob = float(gout) / s.shape[0] if gout.dim() == 0 else None
        if gob is None:
            gob = gout.item() / s.shape[0]
        ds = ext().kd_ce_backward(s, t, gob)
        return ds, None


def kd_soft_cross_entropy(student_logits, teacher_logits):
    """mean_i CE(softmax(teacher_i), student_i) — the KD loss of the
    reference distill example (soft_label cross_entropy,
    example/distill/resnet/train_with_fleet.py:254-259)."""
    if student_logits.is_cuda and available():
        t = teacher_logits.detach()
        if t.dtype != student_logits.dtype:
            t = t.to(student_logits.dtype)
        return _KDSoftCE.apply(student_logits, t)
    # torch reference (CPU tests / numerics baseline)
    logp = torch.log_softmax(student_logits.float(), dim=1)
    soft = torch.softmax(teacher_logits.detach().float(), dim=1)
    return -(soft * logp).sum(1).mean()
