"""FlatCommunicator — one flat-buffer all-reduce per step.

Reference: chainermn/communicators/flat_communicator.py (~L1-70): pack all
grads into one contiguous DeviceMemory buffer, a single all-reduce, unpack
with ×1/size.  Here the pack/unpack are the batched CDNA4 HIP gather/
scatter kernels and the all-reduce is RCCL.
"""

import torch
import torch.distributed as dist

from chainermn_amd import ops
from chainermn_amd.communicators._base import CommunicatorBase
from chainermn_amd.utils.memory import DeviceBuffer


class FlatCommunicator(CommunicatorBase):

    def __init__(self, group=None):
        super().__init__(group=group)
        self._buffer = DeviceBuffer()
        self._plan = None
        self._plan_key = None

    def _get_plan(self, params):
        key = tuple(id(p) for _, p in params)
        if self._plan_key != key:
            device = params[0][1].device if params else self._device
            self._plan = ops.PackPlan([p for _, p in params], device)
            self._plan_key = key
            self._buffer = DeviceBuffer(device=device)
        return self._plan

    def multi_node_mean_grad(self, model, zero_fill=False):
        params = self._grads_for(model, zero_fill)
        if not params:
            return
        plan = self._get_plan(params)
        flat = self._buffer.array(plan.total, dtype=torch.float32)
        ops.pack_grads(plan, flat)
        dist.all_reduce(flat, group=self._group)
        ops.unpack_grads(plan, flat, scale=1.0 / self._size)
