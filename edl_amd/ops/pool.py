"""Pooling on CDNA4 kernels: the vd-shortcut AvgPool2x2 and the stem
MaxPool3x3s2 (parity: the reference model zoo's pool2d ops,
example/distill/resnet/models/resnet_vd.py — avg_pool for the vd
shortcut, 3x3/s2 max_pool after the deep stem).

AvgPool2x2 is a drop-in for nn.AvgPool2d(2, 2, ceil_mode=True) with no
padding: with kernel == stride every input position feeds exactly one
window, so the backward is elementwise (torch's NHWC
avg_pool2d_backward measured 95 us/dispatch). MaxPool3x3s2 keeps u8
tap indices from forward so backward is a <=4-window gather instead of
an argmax re-scan. Plain-torch fallback everywhere else."""
import torch
import torch.nn as nn

from . import available, ext


class _AvgPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.hw = (x.shape[2], x.shape[3])
        return ext().avgpool2x2_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        h, w = ctx.hw
        return ext().avgpool2x2_bwd(dy, h, w)


class AvgPool2x2(nn.Module):
    def __init__(self):
        super().__init__()
        self._fallback = nn.AvgPool2d(2, 2, ceil_mode=True)

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16 and available()
                and x.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            return _AvgPool2x2Fn.apply(x)
        return self._fallback(x)


class _MaxPool3x3s2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = ext().maxpool3x3s2_fwd(x)
        ctx.hw = (x.shape[2], x.shape[3])
        ctx.save_for_backward(idx)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        h, w = ctx.hw
        dy = dy.contiguous(memory_format=torch.channels_last)
        return ext().maxpool3x3s2_bwd(dy.to(torch.bfloat16), idx, h, w)


class MaxPool3x3s2(nn.Module):
    """Drop-in for nn.MaxPool2d(3, 2, padding=1) on NHWC bf16: the fwd
    emits a per-element argmax tap so the backward is a bounded <=4-window
    GATHER (torch's max_pool_backward_nhwc measured 86 us/step at bs32,
    ~0.6 TB/s on its scatter)."""

    def __init__(self):
        super().__init__()
        self._fallback = nn.MaxPool2d(3, 2, padding=1)

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16 and available()
                and x.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            return _MaxPool3x3s2Fn.apply(x)
        return self._fallback(x)
