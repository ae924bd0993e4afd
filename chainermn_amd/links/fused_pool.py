"""FusedMaxPool2d — channels-last 3x3/stride-2/pad-1 max pool on the
gfx950 gather-based kernels (pool.hip).  Drop-in for the ResNet stem's
``nn.MaxPool2d(3, 2, 1)``; any other geometry or layout falls back to
torch's implementation."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from chainermn_amd import ops


class _MaxPool3x3S2(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x):
        y, idx = ops.require().maxpool3x3s2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.in_hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, gy):
        (idx,) = ctx.saved_tensors
        gx = ops.require().maxpool3x3s2_bwd(
            gy.contiguous(memory_format=torch.channels_last), idx,
            ctx.in_hw[0], ctx.in_hw[1])
        return gx


class FusedMaxPool2d(nn.Module):

    def __init__(self, kernel_size=3, stride=2, padding=1):
        super().__init__()
        if (kernel_size, stride, padding) != (3, 2, 1):
            raise ValueError('FusedMaxPool2d supports 3/2/1 only')

    def forward(self, x):
        if x.is_cuda and x.dim() == 4 and x.shape[1] % 8 == 0 \
                and x.is_contiguous(memory_format=torch.channels_last) \
                and not x.is_contiguous():
            return _MaxPool3x3S2.apply(x)
        return F.max_pool2d(x, 3, 2, 1)
