"""Fused BatchNorm (+ReLU, +Add+ReLU) on the CDNA4 BN kernels.

Drop-ins for ``torch.nn.BatchNorm2d`` and for the BN→ReLU /
BN→add-residual→ReLU motifs of ResNet.  Exists because torch's native
channels_last BatchNorm kernels cost 66% of a ResNet-50 training step on
MI355X (see profiles/), and the separate ReLU / residual-add elementwise
kernels cost several more ms of pure HBM re-reads.  The fusions keep the
activation in the producing kernel:

* ``FusedBatchNorm2d``      : y = bn(x)
* ``FusedBNReLU2d``         : y = relu(bn(x)) — backward recomputes the
  relu mask from gamma*xhat+beta (zero extra tensor traffic)
* ``FusedBatchNorm2d.forward_add_relu(x, res)`` :
  y = relu(bn(x) + res) — backward masks with the saved output y and
  emits the residual gradient from the same kernel pass

Kernels: _hip/csrc/mnbn.hip (shared with MultiNodeBatchNormalization).
Running statistics are updated inside the bn_finalize kernel and always
kept fp32 (a bf16-cast model would otherwise hold bf16 stat buffers and
pay a mixed-dtype update kernel per layer per step).
"""

import torch
import torch.nn as nn

from chainermn_amd import ops


def _shape(t, x):
    s = [1] * x.dim()
    s[1] = t.numel()
    return t.view(s)


def _empty(x):
    return x.new_empty(0)


class _FusedBNAct(torch.autograd.Function):
    """relu_mode: 0 = bn, 1 = bn+relu (affine mask), 2 = bn+add+relu.

    ``comm`` non-None → multi-node BN: the (Σx, Σx²) buffer is summed
    across ranks before finalize, and backward all-reduces (Σgy', Σgy'x̂)
    — exact global-batch statistics (assumes rank-equal local batch, as
    the reference did; use scatter_dataset(force_equal_length=True)).
    ``unbias``: running_var correction (n/(n-1) torch-style, or 1.0 for
    the reference's biased convention).
    """

    @staticmethod
    def forward(ctx, x, res, gamma, beta, eps, relu_mode, momentum,
                r_mean, r_var, comm=None, unbias=None, pre_stats=None,
                junction=None, bwd_handoff=None):
        C = x.shape[1]
        count = x.numel() // C
        world = comm.size if comm is not None else 1
        total = count * world
        gamma_f = gamma.to(x.dtype)
        beta_f = beta.to(x.dtype)
        if unbias is None:
            unbias = total / max(total - 1, 1)
        if x.is_cuda:
            ext = ops.require()
            if pre_stats is not None:
                # producer kernel (conv/GEMM epilogue) already reduced
                # (sum y | sum y^2) into [16, 2C] partials — the
                # standalone bn_reduce read pass over x is skipped
                buf = pre_stats.sum(0)
            else:
                buf = torch.empty(2 * C, dtype=torch.float32,
                                  device=x.device)
                ext.bn_stats_into(x, buf)
            if world > 1:
                import torch.distributed as dist
                dist.all_reduce(buf, group=comm.group)
            if r_mean is not None:
                mean, rstd, var = ext.bn_finalize(
                    buf, float(total), eps, momentum, unbias, r_mean, r_var)
            else:
                mean, rstd, var = ext.bn_finalize(
                    buf, float(total), eps, 0.0, 1.0, _empty(buf),
                    _empty(buf))
            y = ext.bn_fwd_act(x, mean, rstd, gamma_f, beta_f,
                               res if res is not None else _empty(x),
                               1 if relu_mode else 0)
        else:
            dims = [d for d in range(x.dim()) if d != 1]
            xf = x.float()
            s_sum = xf.sum(dim=dims)
            sq_sum = (xf * xf).sum(dim=dims)
            if world > 1:
                import torch.distributed as dist
                buf = torch.cat([s_sum, sq_sum])
                dist.all_reduce(buf, group=comm.group)
                s_sum, sq_sum = buf[:C], buf[C:]
            mean = s_sum / total
            var = (sq_sum / total - mean * mean).clamp_min(0)
            rstd = torch.rsqrt(var + eps)
            y = (xf - _shape(mean, x)) * _shape(rstd, x) \
                * _shape(gamma_f.float(), x) + _shape(beta_f.float(), x)
            if res is not None:
                y = y + res.float()
            if relu_mode:
                y = torch.relu(y)
            y = y.to(x.dtype)
            if r_mean is not None:
                with torch.no_grad():
                    r_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                    r_var.mul_(1 - momentum).add_(var * unbias,
                                                  alpha=momentum)
        if relu_mode == 2:
            ctx.save_for_backward(x, gamma, beta, mean, rstd, y)
        else:
            ctx.save_for_backward(x, gamma, beta, mean, rstd)
        ctx.relu_mode = relu_mode
        ctx.count = total
        ctx.comm = comm if world > 1 else None
        ctx.has_res = res is not None
        ctx.junction = junction
        ctx.bwd_handoff = bwd_handoff
        if bwd_handoff is not None and relu_mode == 1 and x.is_cuda:
            # arm the handoff: the conv consuming this BN's output can
            # fuse THIS BN's backward reduce into its dgrad epilogue
            bwd_handoff.update(x=x, mean=mean, rstd=rstd,
                               gamma=gamma_f, beta=beta_f)
        return y

    @staticmethod
    def backward(ctx, gy):
        relu_mode = ctx.relu_mode
        count = ctx.count
        if relu_mode == 2:
            x, gamma, beta, mean, rstd, yout = ctx.saved_tensors
        else:
            x, gamma, beta, mean, rstd = ctx.saved_tensors
            yout = None
        gamma_f = gamma.to(x.dtype)
        beta_f = beta.to(x.dtype)
        C = x.shape[1]
        if x.is_cuda:
            ext = ops.require()
            handoff = getattr(ctx, 'bwd_handoff', None)
            partials = handoff.pop('partials', None) \
                if handoff is not None else None
            if partials is not None:
                # the dgrad kernel that produced gy already reduced
                # (sum g' | sum g'*xhat) in its epilogue
                buf = partials.sum(0)
            else:
                buf = torch.empty(2 * C, dtype=torch.float32,
                                  device=x.device)
                ext.bn_bwd_reduce_act_into(
                    x, gy, mean, rstd, gamma_f, beta_f,
                    yout if yout is not None else _empty(x), relu_mode,
                    buf)
            if ctx.comm is not None:
                import torch.distributed as dist
                dist.all_reduce(buf, group=ctx.comm.group)
            gbeta = buf[:C]
            ggamma = buf[C:]
            gx, gres = ext.bn_bwd_input_act(
                x, gy, mean, rstd, gamma_f, beta_f, gbeta, ggamma,
                yout if yout is not None else _empty(x), relu_mode,
                ctx.has_res, float(count))
        else:
            dims = [d for d in range(x.dim()) if d != 1]
            xhat = (x.float() - _shape(mean, x)) * _shape(rstd, x)
            gyf = gy.float()
            if relu_mode == 1:
                mask = (xhat * _shape(gamma_f.float(), x)
                        + _shape(beta_f.float(), x)) > 0
                gyf = gyf * mask
            elif relu_mode == 2:
                gyf = gyf * (yout.float() > 0)
            gbeta = gyf.sum(dim=dims)
            ggamma = (gyf * xhat).sum(dim=dims)
            if ctx.comm is not None:
                import torch.distributed as dist
                buf = torch.cat([gbeta, ggamma])
                dist.all_reduce(buf, group=ctx.comm.group)
                gbeta, ggamma = buf[:C], buf[C:]
            gx = (_shape(gamma_f.float() * rstd, x)
                  * (gyf - (xhat * _shape(ggamma, x)
                            + _shape(gbeta, x)) / count)).to(x.dtype)
            gres = gyf.to(x.dtype) if ctx.has_res else None
        g_res = gres if ctx.has_res else None
        if ctx.junction is not None and g_res is not None:
            # residual-junction handoff: the SAME tensor autograd routes
            # to _ResidualEntry's identity slot; the block's conv1 dgrad
            # may accumulate the main-path gradient into it in-kernel
            ctx.junction['gres'] = g_res
        return (gx, g_res, ggamma.to(gamma.dtype), gbeta.to(beta.dtype),
                None, None, None, None, None, None, None, None, None,
                None)


class FusedBatchNorm2d(nn.Module):
    """torch.nn.BatchNorm2d-compatible (momentum/eps/affine semantics,
    unbiased running_var) on the fused gfx950 kernels."""

    _relu_mode = 0
    comm = None            # set → global-batch (multi-node) statistics
    _running_unbias = None  # None = torch n/(n-1); 1.0 = biased (chainermn)

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

    def _apply(self, fn, recurse=True):
        # num_batches_tracked stays a CPU scalar: incrementing a GPU long
        # tensor costs one kernel launch per layer per step
        ret = super()._apply(fn, recurse)
        if self.num_batches_tracked is not None \
                and self.num_batches_tracked.is_cuda:
            self.num_batches_tracked.data = \
                self.num_batches_tracked.data.cpu()
        return ret

    def _gamma_beta(self, x):
        gamma = self.weight if self.weight is not None else torch.ones(
            self.num_features, device=x.device, dtype=x.dtype)
        beta = self.bias if self.bias is not None else torch.zeros(
            self.num_features, device=x.device, dtype=x.dtype)
        return gamma, beta

    def _running(self):
        if not self.track_running_stats:
            return None, None, 0.0
        if self.running_mean.dtype != torch.float32:
            # keep stats fp32 even when the model is cast to bf16
            self.running_mean.data = self.running_mean.data.float()
            self.running_var.data = self.running_var.data.float()
        m = self.momentum if self.momentum is not None else 0.1
        return self.running_mean, self.running_var, m

    def _train_forward(self, x, res, relu_mode, pre_stats=None,
                       junction=None, bwd_handoff=None):
        gamma, beta = self._gamma_beta(x)
        r_mean, r_var, m = self._running()
        y = _FusedBNAct.apply(x, res, gamma, beta, self.eps, relu_mode, m,
                              r_mean, r_var, self.comm,
                              self._running_unbias, pre_stats, junction,
                              bwd_handoff)
        if self.track_running_stats:
            self.num_batches_tracked += 1
        return y

    def _eval_forward(self, x, res, relu_mode):
        gamma, beta = self._gamma_beta(x)
        rstd = torch.rsqrt(self.running_var.float() + self.eps)
        if x.is_cuda:
            return ops.require().bn_fwd_act(
                x, self.running_mean.float(), rstd, gamma.to(x.dtype),
                beta.to(x.dtype), res if res is not None else x.new_empty(0),
                1 if relu_mode else 0)
        y = (x.float() - _shape(self.running_mean.float(), x)) \
            * _shape(rstd, x) * _shape(gamma.float(), x) \
            + _shape(beta.float(), x)
        if res is not None:
            y = y + res.float()
        if relu_mode:
            y = torch.relu(y)
        return y.to(x.dtype)

    def forward(self, x, pre_stats=None, bwd_handoff=None):
        if self.training or not self.track_running_stats:
            return self._train_forward(x, None, self._relu_mode,
                                       pre_stats,
                                       bwd_handoff=bwd_handoff)
        return self._eval_forward(x, None, self._relu_mode)

    def forward_add_relu(self, x, res, pre_stats=None, junction=None):
        """y = relu(bn(x) + res) in one kernel pass (ResNet residual)."""
        if self.training or not self.track_running_stats:
            return self._train_forward(x, res, 2, pre_stats, junction)
        return self._eval_forward(x, res, 2)

    def extra_repr(self):
        return '{num_features}, eps={eps}, momentum={momentum}, ' \
               'affine={affine}'.format(**self.__dict__)


class FusedBNReLU2d(FusedBatchNorm2d):
    """y = relu(bn(x)) — the relu never touches HBM separately."""

    _relu_mode = 1
