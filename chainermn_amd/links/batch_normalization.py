"""MultiNodeBatchNormalization module.

Reference: chainermn/links/batch_normalization.py (~L1-180): drop-in for
the framework's BatchNormalization that computes batch statistics over the
*global* (cross-rank) batch and maintains running statistics from them.
Essential for large-batch training (BASELINE.json config 5: batch 8192 +
LARS).

Implementation: the fused CDNA4 BN engine (fused_batch_norm._FusedBNAct)
with a communicator attached — the (Σx, Σx², Σgy', Σgy'x̂) buffers go over
RCCL between the local reduction and the finalize/input kernels.  Running
statistics use the reference's convention (decay; unbiased m/(m-1)
running-variance adjust, with m = the global element count).
``MultiNodeBNReLU2d`` / ``forward_add_relu`` give the multi-node BN the
same ReLU / residual fusions as the local FusedBatchNorm2d.
"""

import torch
import torch.nn as nn

from chainermn_amd.links.fused_batch_norm import FusedBatchNorm2d


class MultiNodeBatchNormalization(FusedBatchNorm2d):

    # Upstream Chainer/ChainerMN BN applies the unbiased m/(m-1) adjust
    # when folding batch var into running_var; None selects the engine's
    # n/(n-1) with n = the GLOBAL element count (the statistics the var
    # was computed over), matching that convention.
    _running_unbias = None

    def __init__(self, size, comm, decay=0.9, eps=2e-5, dtype=None,
                 use_gamma=True, use_beta=True, communication_backend='auto'):
        if communication_backend not in ('auto', 'rccl', 'nccl', 'gloo'):
            raise ValueError(
                'communication_backend must be auto/rccl/nccl/gloo')
        super().__init__(size, eps=eps, momentum=1.0 - decay,
                         affine=True, track_running_stats=True)
        self.comm = comm
        self.decay = decay
        dtype = dtype or torch.float32
        if use_gamma:
            self.weight = nn.Parameter(torch.ones(size, dtype=dtype))
        else:
            self.register_parameter('weight', None)
        if use_beta:
            self.bias = nn.Parameter(torch.zeros(size, dtype=dtype))
        else:
            self.register_parameter('bias', None)

    def extra_repr(self):
        return '{}, decay={}, eps={}, comm_size={}'.format(
            self.num_features, self.decay, self.eps, self.comm.size)


class MultiNodeBNReLU2d(MultiNodeBatchNormalization):
    """y = relu(global-batch-bn(x)) in one kernel pass."""

    _relu_mode = 1
