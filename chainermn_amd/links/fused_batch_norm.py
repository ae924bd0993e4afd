"""FusedBatchNorm2d — local (per-rank) BatchNorm on the CDNA4 BN kernels.

Drop-in for ``torch.nn.BatchNorm2d``.  Exists because torch's native
channels_last BatchNorm kernels cost 66% of a ResNet-50 training step on
MI355X (see profiles/): its collect_statistics/backward_reduce kernels run
~20x above the bandwidth bound.  This module runs the same four fused
passes as MultiNodeBatchNormalization (bn_stats / bn_fwd / bn_bwd_reduce /
bn_bwd_input in _hip/csrc/mnbn.hip) without the cross-rank all-reduce —
the data-parallel configs of the reference used per-rank local BN
(global-batch BN is the separate MNBN link).
"""

import torch
import torch.nn as nn

from chainermn_amd import ops


def _shape(t, x):
    s = [1] * x.dim()
    s[1] = t.numel()
    return t.view(s)


class _FusedBatchNorm(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, gamma, beta, eps, momentum=0.0, r_mean=None,
                r_var=None):
        if x.is_cuda:
            ext = ops.require()
            C = x.shape[1]
            count = x.numel() // C
            buf = torch.empty(2 * C, dtype=torch.float32, device=x.device)
            ext.bn_stats_into(x, buf)
            if r_mean is not None:
                unbias = count / max(count - 1, 1)
                mean, rstd, var = ext.bn_finalize(
                    buf, float(count), eps, momentum, unbias, r_mean, r_var)
            else:
                mean, rstd, var = ext.bn_finalize(buf, float(count), eps, 0.0,
                                                  1.0, buf.new_empty(0),
                                                  buf.new_empty(0))
            y = ext.bn_fwd(x, mean, rstd, gamma.float(), beta.float())
        else:
            dims = [d for d in range(x.dim()) if d != 1]
            xf = x.float()
            count = x.numel() // x.shape[1]
            mean = xf.mean(dim=dims)
            var = xf.var(dim=dims, unbiased=False)
            rstd = torch.rsqrt(var + eps)
            y = ((xf - _shape(mean, x)) * _shape(rstd, x)
                 * _shape(gamma.float(), x)
                 + _shape(beta.float(), x)).to(x.dtype)
        ctx.save_for_backward(x, gamma, mean, rstd)
        ctx.count = count
        return y, mean, var

    @staticmethod
    def backward(ctx, gy, g_mean, g_var):
        x, gamma, mean, rstd = ctx.saved_tensors
        count = ctx.count
        if x.is_cuda:
            ext = ops.require()
            C = x.shape[1]
            buf = torch.empty(2 * C, dtype=torch.float32, device=x.device)
            ext.bn_bwd_reduce_into(x, gy, mean, rstd, buf)
            gbeta = buf[:C]
            ggamma = buf[C:]
            gx = ext.bn_bwd_input(x, gy, mean, rstd, gamma.float(), gbeta,
                                  ggamma, float(count))
        else:
            dims = [d for d in range(x.dim()) if d != 1]
            gyf = gy.float()
            xhat = (x.float() - _shape(mean, x)) * _shape(rstd, x)
            gbeta = gyf.sum(dim=dims)
            ggamma = (gyf * xhat).sum(dim=dims)
            gx = (_shape(gamma.float() * rstd, x)
                  * (gyf - (xhat * _shape(ggamma, x)
                            + _shape(gbeta, x)) / count)).to(x.dtype)
        return (gx, ggamma.to(gamma.dtype), gbeta.to(gamma.dtype),
                None, None, None, None)


class FusedBatchNorm2d(nn.Module):
    """torch.nn.BatchNorm2d-compatible (momentum/eps/affine semantics,
    unbiased running_var) on the fused gfx950 kernels."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.affine = affine
        self.track_running_stats = track_running_stats
        if affine:
            self.weight = nn.Parameter(torch.ones(num_features))
            self.bias = nn.Parameter(torch.zeros(num_features))
        else:
            self.register_parameter('weight', None)
            self.register_parameter('bias', None)
        if track_running_stats:
            self.register_buffer('running_mean', torch.zeros(num_features))
            self.register_buffer('running_var', torch.ones(num_features))
            self.register_buffer('num_batches_tracked',
                                 torch.tensor(0, dtype=torch.long))
        else:
            self.register_buffer('running_mean', None)
            self.register_buffer('running_var', None)

    def forward(self, x):
        gamma = self.weight if self.weight is not None else torch.ones(
            self.num_features, device=x.device, dtype=x.dtype)
        beta = self.bias if self.bias is not None else torch.zeros(
            self.num_features, device=x.device, dtype=x.dtype)
        if self.training or not self.track_running_stats:
            if self.track_running_stats and x.is_cuda:
                # running stats stay fp32 even when the model is cast to
                # bf16 (a bf16 buffer would force mixed-dtype update
                # kernels AND destroy the stats' precision)
                if self.running_mean.dtype != torch.float32:
                    self.running_mean.data = self.running_mean.data.float()
                    self.running_var.data = self.running_var.data.float()
                m = self.momentum if self.momentum is not None else 0.1
                y, mean, var = _FusedBatchNorm.apply(
                    x, gamma, beta, self.eps, m, self.running_mean,
                    self.running_var)
                with torch.no_grad():
                    self.num_batches_tracked += 1
                return y
            y, mean, var = _FusedBatchNorm.apply(x, gamma, beta, self.eps)
            if self.track_running_stats:
                with torch.no_grad():
                    n = x.numel() // x.shape[1]
                    m = self.momentum if self.momentum is not None else 0.1
                    rm = self.running_mean
                    if rm.dtype != torch.float32:
                        self.running_mean.data = rm.data.float()
                        self.running_var.data = self.running_var.data.float()
                    self.running_mean.mul_(1 - m).add_(mean.detach(),
                                                       alpha=m)
                    unbiased = var.detach() * (n / max(n - 1, 1))
                    self.running_var.mul_(1 - m).add_(unbiased, alpha=m)
                    self.num_batches_tracked += 1
            return y
        # eval mode: normalize with running stats
        rstd = torch.rsqrt(self.running_var.float() + self.eps)
        if x.is_cuda:
            return ops.require().bn_fwd(x, self.running_mean.float(), rstd,
                                        gamma.float(), beta.float())
        y = ((x.float() - _shape(self.running_mean, x)) * _shape(rstd, x)
             * _shape(gamma.float(), x) + _shape(beta.float(), x))
        return y.to(x.dtype)

    def extra_repr(self):
        return '{num_features}, eps={eps}, momentum={momentum}, ' \
               'affine={affine}'.format(**self.__dict__)
