"""NaiveCommunicator — per-parameter all-reduce.

Reference: chainermn/communicators/naive_communicator.py (~L1-50): one
MPI_Allreduce per parameter, then divide by size.  Here: one
torch.distributed all_reduce per parameter (gloo on CPU, RCCL on GPU).
The only communicator guaranteed to work with CPU-only worlds — used by
the MNIST/plumbing config and the CPU test suite.
"""

import torch.distributed as dist

from chainermn_amd.communicators._base import CommunicatorBase


class NaiveCommunicator(CommunicatorBase):

    def multi_node_mean_grad(self, model, zero_fill=False):
        scale = 1.0 / self._size
        for _, p in self._grads_for(model, zero_fill):
            dist.all_reduce(p.grad, group=self._group)
            p.grad.mul_(scale)
