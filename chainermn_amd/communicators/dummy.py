"""DummyCommunicator — single-process stand-in, no collectives.

Reference: chainermn/communicators/dummy_communicator.py (~L1-40).
Used by tests and single-GPU debugging without a launcher.
"""

import torch

from chainermn_amd.communicators._base import CommunicatorBase


class DummyCommunicator(CommunicatorBase):

    def __init__(self, group=None):
        # No process group at all — geometry is fixed at 1.
        self._group = None
        self._rank = 0
        self._size = 1
        self._intra_rank = 0
        self._intra_size = 1
        self._inter_rank = 0
        self._inter_size = 1
        self._local_ranks = [0]
        self._device = torch.device('cpu')
        if torch.cuda.is_available():
            self._device = torch.device('cuda', torch.cuda.current_device())

    def multi_node_mean_grad(self, model, zero_fill=False):
        self._grads_for(model, zero_fill)  # materialise zero_fill grads only

    def bcast_data(self, model, root=0):
        pass

    def send_obj(self, obj, dest, tag=0, max_buf_len=None):
        raise RuntimeError('DummyCommunicator has no peers')

    def recv_obj(self, source, tag=0):
        raise RuntimeError('DummyCommunicator has no peers')

    def bcast_obj(self, obj, root=0):
        return obj

    def gather_obj(self, obj, root=0):
        return [obj]

    def allgather_obj(self, obj):
        return [obj]

    def scatter_obj(self, objs, root=0, max_buf_len=None):
        return objs[0]

    def allreduce_obj(self, obj):
        return obj

    def bcast(self, x, root=0):
        return x

    def allreduce(self, x, op='sum'):
        return x.detach().clone()

    def allgather(self, x):
        return (x.detach().clone(),)

    def alltoall(self, xs):
        return (xs[0].detach().clone(),)

    def gather(self, x, root=0):
        return (x.detach().clone(),)

    def scatter(self, xs, root=0):
        return xs[0].detach().clone()

    def split(self, color, key=None):
        return DummyCommunicator()

    def barrier(self):
        pass
