"""MultiNodeBatchNormalization module.

Reference: chainermn/links/batch_normalization.py (~L1-180): drop-in for
the framework's BatchNormalization that computes batch statistics over the
*global* (cross-rank) batch and maintains running statistics from them.
Essential for large-batch training (BASELINE.json config 5: batch 8192 +
LARS).  Here it is a drop-in for ``torch.nn.BatchNorm1d/2d/3d``.
"""

import torch
import torch.nn as nn

from chainermn_amd.functions.batch_normalization import multi_node_batch_norm


class MultiNodeBatchNormalization(nn.Module):

    def __init__(self, size, comm, decay=0.9, eps=2e-5, dtype=None,
                 use_gamma=True, use_beta=True, communication_backend='auto'):
        super().__init__()
        if communication_backend not in ('auto', 'rccl', 'nccl', 'gloo'):
            raise ValueError(
                'communication_backend must be auto/rccl/nccl/gloo')
        self.comm = comm
        self.eps = eps
        self.decay = decay  # torch momentum = 1 - decay
        self.num_features = size
        dtype = dtype or torch.float32
        if use_gamma:
            self.weight = nn.Parameter(torch.ones(size, dtype=dtype))
        else:
            self.register_parameter('weight', None)
        if use_beta:
            self.bias = nn.Parameter(torch.zeros(size, dtype=dtype))
        else:
            self.register_parameter('bias', None)
        self.register_buffer('running_mean', torch.zeros(size))
        self.register_buffer('running_var', torch.ones(size))
        self.register_buffer('num_batches_tracked',
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x):
        if self.training:
            gamma = (self.weight if self.weight is not None
                     else torch.ones(self.num_features, device=x.device,
                                     dtype=x.dtype))
            beta = (self.bias if self.bias is not None
                    else torch.zeros(self.num_features, device=x.device,
                                     dtype=x.dtype))
            y, mean, var = multi_node_batch_norm(x, gamma, beta, self.comm,
                                                 self.eps)
            with torch.no_grad():
                m = 1.0 - self.decay
                self.running_mean.mul_(self.decay).add_(mean.detach(),
                                                        alpha=m)
                self.running_var.mul_(self.decay).add_(var.detach(), alpha=m)
                self.num_batches_tracked += 1
            return y
        else:
            shape = [1] * x.dim()
            shape[1] = self.num_features
            rstd = torch.rsqrt(self.running_var + self.eps)
            y = (x.float() - self.running_mean.view(shape)) \
                * rstd.view(shape)
            if self.weight is not None:
                y = y * self.weight.float().view(shape)
            if self.bias is not None:
                y = y + self.bias.float().view(shape)
            return y.to(x.dtype)

    def extra_repr(self):
        return '{}, decay={}, eps={}, comm_size={}'.format(
            self.num_features, self.decay, self.eps, self.comm.size)
