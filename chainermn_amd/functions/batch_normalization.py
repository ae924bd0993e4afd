"""Multi-node batch normalization — the autograd function.

Reference: chainermn/functions/batch_normalization.py (~L1-170): forward
computes local per-channel mean / mean-of-squares (2×C floats), all-reduces
them into global statistics and normalizes with those; backward all-reduces
the local (gbeta, ggamma) pair and applies the standard BN input-gradient
formula with the *global* statistics.

MI355X: the local reductions and the normalize are hand-written CDNA4 HIP
kernels (fp32 accumulation — see _hip/csrc/mnbn.hip); the 2C+1-element
statistics buffer goes over RCCL.  CPU tensors use torch reference ops
(exercised by the CPU test suite as the numerics oracle).
"""

import torch

from chainermn_amd import ops


def _local_sums(x):
    """(sum_c, sqsum_c, count) reduced over every dim but dim 1 (CPU)."""
    dims = [d for d in range(x.dim()) if d != 1]
    xf = x.float()
    return (xf.sum(dim=dims), (xf * xf).sum(dim=dims),
            x.numel() // x.shape[1])


def _bcast_shape(t, x):
    shape = [1] * x.dim()
    shape[1] = t.numel()
    return t.view(shape)


class _MultiNodeBatchNorm(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, gamma, beta, comm, eps):
        C = x.shape[1]
        buf = torch.empty(2 * C + 1, dtype=torch.float32, device=x.device)
        if x.is_cuda:
            ext = ops.require()
            ext.bn_stats_into(x, buf)
            buf[2 * C] = float(x.numel() // C)
            if comm.size > 1:
                buf = comm.allreduce(buf)
            mean, rstd, var = ext.bn_finalize(buf, -1.0, eps, 0.0, 1.0,
                                              buf.new_empty(0),
                                              buf.new_empty(0))
            total = buf[2 * C]
        else:
            s, sq, count = _local_sums(x)
            buf[:C] = s
            buf[C:2 * C] = sq
            buf[2 * C] = float(count)
            buf = comm.allreduce(buf)
            total = buf[2 * C]
            mean = buf[:C] / total
            var = buf[C:2 * C] / total - mean * mean
            rstd = torch.rsqrt(var.clamp_min(0) + eps)
        if x.is_cuda:
            y = ops.require().bn_fwd(x, mean, rstd, gamma.to(x.dtype),
                                     beta.to(x.dtype))
        else:
            xhat = (x.float() - _bcast_shape(mean, x)) * _bcast_shape(rstd, x)
            y = (xhat * _bcast_shape(gamma.float(), x)
                 + _bcast_shape(beta.float(), x)).to(x.dtype)
        ctx.save_for_backward(x, gamma, mean, rstd, total)
        ctx.comm = comm
        # The returned statistics are a by-product: backward ignores
        # g_mean/g_var, so a user graph built through them would silently
        # get zero gradient.  Marking them non-differentiable makes such
        # misuse raise instead.
        ctx.mark_non_differentiable(mean, var)
        return y, mean, var

    @staticmethod
    def backward(ctx, gy, g_mean, g_var):
        x, gamma, mean, rstd, total = ctx.saved_tensors
        comm = ctx.comm
        C = x.shape[1]
        buf = torch.empty(2 * C, dtype=torch.float32, device=x.device)
        if x.is_cuda:
            ops.require().bn_bwd_reduce_into(x, gy, mean, rstd, buf)
        else:
            dims = [d for d in range(x.dim()) if d != 1]
            gyf = gy.float()
            xhat = (x.float() - _bcast_shape(mean, x)) * _bcast_shape(rstd, x)
            buf[:C] = gyf.sum(dim=dims)
            buf[C:] = (gyf * xhat).sum(dim=dims)
        if comm.size > 1:
            buf = comm.allreduce(buf)
        gbeta_g = buf[:C]
        ggamma_g = buf[C:]
        if x.is_cuda:
            gx = ops.require().bn_bwd_input(x, gy, mean, rstd,
                                            gamma.to(x.dtype),
                                            gbeta_g, ggamma_g, float(total))
        else:
            gyf = gy.float()
            xhat = (x.float() - _bcast_shape(mean, x)) * _bcast_shape(rstd, x)
            coef = _bcast_shape(gamma.float() * rstd, x)
            gx = (coef * (gyf - (xhat * _bcast_shape(ggamma_g, x)
                                 + _bcast_shape(gbeta_g, x)) / total)
                  ).to(x.dtype)
        return (gx, ggamma_g.to(gamma.dtype), gbeta_g.to(gamma.dtype),
                None, None)


def multi_node_batch_norm(x, gamma, beta, comm, eps=2e-5):
    """Functional multi-node BN; returns (y, global_mean, global_var)."""
    return _MultiNodeBatchNorm.apply(x, gamma, beta, comm, eps)
