"""FusedConv1x1 — 1x1 convolutions as hipBLASLt GEMMs.

A channels-last 1x1/stride-1 convolution IS a GEMM over
[N*H*W, Cin] x [Cin, Cout] (the permuted view of a channels_last tensor is
free).  Measured on MI355X (batch 512, see profiles/): hipBLASLt GEMMs run
1.4-1.6x faster than MIOpen's igemm conv kernels for ResNet-50's 1x1
shapes on forward and data-grad.  The weight-grad GEMM shape
([Cout, M] x [M, Cin], huge K, tiny output) is pathological for a plain
GEMM call, so backward routes grad_weight through MIOpen's tuned
split-K wrw kernels via aten::convolution_backward.
"""

import torch
import torch.nn as nn

from chainermn_amd import ops


class _Conv1x1Fn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, w):
        # x: channels_last [N, C, H, W]; its (0,2,3,1) permute is a free
        # contiguous view
        N, C, H, W = x.shape
        x2 = x.permute(0, 2, 3, 1).reshape(-1, C)
        w2 = w.reshape(w.shape[0], C)
        y2 = ops.gemm_bt(x2, w2)  # autotuned: MFMA kernel vs hipBLASLt
        y = y2.view(N, H, W, -1).permute(0, 3, 1, 2)
        ctx.save_for_backward(x, w)
        return y

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        N, C, H, W = x.shape
        Cout = w.shape[0]
        gy = gy.contiguous(memory_format=torch.channels_last)
        gy2 = gy.permute(0, 2, 3, 1).reshape(-1, Cout)
        w2 = w.reshape(Cout, C)
        gx2 = ops.gemm_nn(gy2, w2)  # autotuned
        gx = gx2.view(N, H, W, C).permute(0, 3, 1, 2)
        # weight grad: MIOpen's tuned split-K wrw kernels.  (The plain
        # GEMM form [Cout, M] x [M, Cin] with M ~ 1e6 is ~10x slower; a
        # hipBLASLt bmm split-K looked 8-30% faster in isolation but
        # measured 2 ms/step SLOWER end to end — within-box A/B in
        # profiles/README.md — so MIOpen stays.)
        import os
        if os.environ.get('CHAINERMN_AMD_WGRAD') == 'bmm':
            M = gy2.shape[0]
            x2 = x.permute(0, 2, 3, 1).reshape(M, C)
            S = next((s for s in (128, 64, 32, 16, 8, 4, 2, 1)
                      if M % s == 0))
            part = torch.bmm(gy2.view(S, M // S, Cout).transpose(1, 2),
                             x2.view(S, M // S, C))
            gw = part.float().sum(0).to(w.dtype).view_as(w)
            return gx, gw
        _, gw, _ = torch.ops.aten.convolution_backward(
            gy, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
            [False, True, False])
        return gx, gw


class FusedConv1x1(nn.Module):
    """Drop-in for nn.Conv2d(cin, cout, 1, bias=False) on channels_last."""

    def __init__(self, in_channels, out_channels):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        # identical initialization (and RNG consumption) to
        # nn.Conv2d(cin, cout, 1, bias=False)
        ref = nn.Conv2d(in_channels, out_channels, 1, bias=False)
        self.weight = ref.weight

    def forward(self, x):
        if x.is_cuda and x.dim() == 4 \
                and x.is_contiguous(memory_format=torch.channels_last) \
                and not x.is_contiguous():
            return _Conv1x1Fn.apply(x, self.weight)
        return torch.nn.functional.conv2d(x, self.weight)

    def extra_repr(self):
        return '{}, {}, kernel_size=1 (GEMM)'.format(self.in_channels,
                                                     self.out_channels)


class _Conv3x3Fn(torch.autograd.Function):
    """3x3/s1/p1 conv on the phased implicit-GEMM MFMA kernel.

    dgrad REUSES the forward kernel: gx = conv3x3s1p1(gy, W^T with taps
    flipped) — the transpose/flip is a 4.7 MB-max copy per backward,
    trivial next to the conv.  wgrad stays on MIOpen's split-K wrw
    kernels (same reasoning as _Conv1x1Fn)."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        return ops.conv3x3(x, w)

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        gy = gy.contiguous(memory_format=torch.channels_last)
        wt = w.flip(2, 3).permute(1, 0, 2, 3).contiguous(
            memory_format=torch.channels_last)
        gx = ops.conv3x3(gy, wt)
        gw = ops.conv3x3_wgrad(x, gy, w)
        return gx, gw


class FusedConv3x3(nn.Module):
    """Drop-in for nn.Conv2d(cin, cout, 3, stride, padding=1, bias=False).

    stride-1 channels_last bf16 inputs run the hand-written CDNA4
    implicit-GEMM kernel (autotuned vs MIOpen per shape); anything else
    (stride 2, CPU, fp32) falls back to F.conv2d."""

    def __init__(self, in_channels, out_channels, stride=1):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.stride = stride
        ref = nn.Conv2d(in_channels, out_channels, 3, stride=stride,
                        padding=1, bias=False)
        self.weight = ref.weight

    def forward(self, x):
        if (self.stride == 1 and x.is_cuda and x.dim() == 4
                and x.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and not x.is_contiguous()):
            return _Conv3x3Fn.apply(x, self.weight)
        return torch.nn.functional.conv2d(x, self.weight,
                                          stride=self.stride, padding=1)

    def extra_repr(self):
        return '{}, {}, kernel_size=3, stride={} (implicit GEMM)'.format(
            self.in_channels, self.out_channels, self.stride)


class _Conv1x1StatsFn(torch.autograd.Function):
    """_Conv1x1Fn variant that also returns the producer-fused BN stats
    partials (empty tensor when the dispatched kernel can't fuse them).

    ``junction`` (a dict or None) is the residual-junction handoff: when
    the SAME block's bn3 backward has deposited the identity-path
    gradient there, this conv's dgrad GEMM accumulates into that tensor
    in its epilogue (ops.gemm_nn_accum) and marks the junction 'fused' —
    _ResidualEntry.backward then skips the engine's separate add."""

    @staticmethod
    def forward(ctx, x, w, junction=None, bwd_bn=None):
        N, C, H, W = x.shape
        x2 = x.permute(0, 2, 3, 1).reshape(-1, C)
        w2 = w.reshape(w.shape[0], C)
        y2, stats = ops.gemm_bt_with_stats(x2, w2)
        y = y2.view(N, H, W, -1).permute(0, 3, 1, 2)
        ctx.save_for_backward(x, w)
        ctx.junction = junction
        ctx.bwd_bn = bwd_bn
        if stats is None:
            stats = x.new_empty(0, dtype=torch.float32)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    def backward(ctx, gy, g_stats):
        # junction and bwd_bn are mutually exclusive by construction
        # (Bottleneck arms junction on conv1, bwd_bn on conv3)
        bnh = getattr(ctx, 'bwd_bn', None)
        if bnh and 'x' in bnh:
            # 1x1 dgrad with the downstream BN's backward reduce fused
            x, w = ctx.saved_tensors
            N, C, H, W = x.shape
            Cout = w.shape[0]
            gy = gy.contiguous(memory_format=torch.channels_last)
            gy2 = gy.permute(0, 2, 3, 1).reshape(-1, Cout)
            w2 = w.reshape(Cout, C)
            gx2, partials = ops.gemm_nn_bnbwd(gy2, w2, bnh)
            if partials is not None:
                bnh['partials'] = partials
            gx = gx2.view(N, H, W, C).permute(0, 3, 1, 2)
            _, gw, _ = torch.ops.aten.convolution_backward(
                gy, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0],
                1, [False, True, False])
            return gx, gw, None, None
        j = ctx.junction
        gres = j.pop('gres', None) if j is not None else None
        if gres is not None:
            x, w = ctx.saved_tensors
            N, C, H, W = x.shape
            Cout = w.shape[0]
            gy = gy.contiguous(memory_format=torch.channels_last)
            gy2 = gy.permute(0, 2, 3, 1).reshape(-1, Cout)
            w2 = w.reshape(Cout, C)
            gres2 = gres.permute(0, 2, 3, 1).reshape(-1, C)
            if gres2.is_contiguous():
                res2, folded = ops.gemm_nn_accum(gy2, w2, gres2)
            else:
                res2, folded = ops.gemm_nn(gy2, w2), False
            if folded:
                j['fused'] = True
                gx = gres
            else:
                gx = res2.view(N, H, W, C).permute(0, 3, 1, 2)
            _, gw, _ = torch.ops.aten.convolution_backward(
                gy, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0],
                1, [False, True, False])
            return gx, gw, None, None
        gx, gw = _Conv1x1Fn.backward(ctx, gy)
        return gx, gw, None, None


class _ResidualEntry(torch.autograd.Function):
    """Explicit fork of a residual block's input: backward adds the two
    path gradients itself UNLESS the conv1-dgrad already folded the
    main-path gradient into the identity gradient (junction['fused'])."""

    @staticmethod
    def forward(ctx, x, junction):
        ctx.junction = junction
        return x, x

    @staticmethod
    def backward(ctx, g_main, g_id):
        j = ctx.junction
        if j is not None and j.pop('fused', False):
            return g_main, None  # g_id was folded into g_main in-kernel
        if g_id is None:
            return g_main, None
        if g_main is None:
            return g_id, None
        return g_main + g_id, None


class _Conv3x3StatsFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, w, bwd_bn=None):
        ctx.save_for_backward(x, w)
        ctx.bwd_bn = bwd_bn
        y, stats = ops.conv3x3_with_stats(x, w)
        if stats is None:
            stats = x.new_empty(0, dtype=torch.float32)
        ctx.mark_non_differentiable(stats)
        return y, stats

    @staticmethod
    def backward(ctx, gy, g_stats):
        bnh = getattr(ctx, 'bwd_bn', None)
        if bnh and 'x' in bnh:
            x, w = ctx.saved_tensors
            gy = gy.contiguous(memory_format=torch.channels_last)
            wt = w.flip(2, 3).permute(1, 0, 2, 3).contiguous(
                memory_format=torch.channels_last)
            gx, partials = ops.conv3x3_with_bnbwd(gy, wt, bnh)
            if partials is not None:
                bnh['partials'] = partials
            gw = ops.conv3x3_wgrad(x, gy, w)
            return gx, gw, None
        gx, gw = _Conv3x3Fn.backward(ctx, gy)
        return gx, gw, None


def _stats_or_none(st):
    return st if st.numel() else None


# forward_stats is attached to the module classes here (rather than in
# the class bodies above) so the Function variants it dispatches to can
# be defined after the base classes they reuse.


def _conv1x1_forward_stats(self, x, junction=None, bwd_bn=None):
    """(y, BN-stats partials or None) — the fused conv->bn fast path."""
    if x.is_cuda and x.dim() == 4 \
            and x.is_contiguous(memory_format=torch.channels_last) \
            and not x.is_contiguous():
        y, st = _Conv1x1StatsFn.apply(x, self.weight, junction, bwd_bn)
        return y, _stats_or_none(st)
    return self.forward(x), None


FusedConv1x1.forward_stats = _conv1x1_forward_stats


def _conv3x3_forward_stats(self, x, bwd_bn=None):
    """(y, BN-stats partials or None) — the fused conv->bn fast path."""
    if (self.stride == 1 and x.is_cuda and x.dim() == 4
            and x.dtype == torch.bfloat16
            and x.is_contiguous(memory_format=torch.channels_last)
            and not x.is_contiguous()):
        y, st = _Conv3x3StatsFn.apply(x, self.weight, bwd_bn)
        return y, _stats_or_none(st)
    return self.forward(x), None


FusedConv3x3.forward_stats = _conv3x3_forward_stats
