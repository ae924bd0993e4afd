"""PureRcclCommunicator — the flagship gradient path.

Reference: chainermn/communicators/pure_nccl_communicator.py (~L1-340):
pack grads → optional cast to ``allreduce_grad_dtype`` (fp16) → one
ncclAllReduce over all ranks on a dedicated stream → fused cast-back +
×1/size → unpack, with two growable DeviceMemory regions.

MI355X redesign (not a translation):

* The compressed dtype defaults to **bf16** (same wire bytes as fp16,
  fp32-range exponents, native CDNA4) rather than fp16.
* Pack+cast and unpack+scale+cast-back are single fused CDNA4 HIP kernels
  (one HBM round-trip each, gfx950 — see _hip/csrc/pack.hip).
* The all-reduce is RCCL over xGMI (7 point-to-point links/GPU); an
  ``async_`` entry point runs pack+all-reduce on a caller-owned side HIP
  stream so the double-buffering optimizer can overlap it with
  forward/backward (reference's _DoubleBufferingOptimizer thread+stream,
  optimizers.py ~L120-300).
"""

import torch
import torch.distributed as dist

from chainermn_amd import ops
from chainermn_amd.communicators._base import CommunicatorBase
from chainermn_amd.utils.memory import DeviceBuffer

_DTYPES = {None: None, 'fp32': None, torch.float32: None,
           'bf16': torch.bfloat16, torch.bfloat16: torch.bfloat16,
           'fp16': torch.float16, torch.float16: torch.float16}


class PureRcclCommunicator(CommunicatorBase):

    def __init__(self, group=None, allreduce_grad_dtype='bf16',
                 batched_copy=True):
        super().__init__(group=group)
        if allreduce_grad_dtype not in _DTYPES:
            raise ValueError(
                'allreduce_grad_dtype must be one of None/fp32/bf16/fp16')
        self.allreduce_grad_dtype = _DTYPES[allreduce_grad_dtype]
        # batched_copy kept for reference API parity; the batched kernels
        # are always used on GPU (the per-param cudaMemcpy loop the flag
        # guarded in the reference has no reason to exist here).
        self.batched_copy = batched_copy
        self._buf_a = DeviceBuffer()
        self._buf_b = DeviceBuffer()
        self._plan = None
        self._plan_key = None

    @classmethod
    def _from_group(cls, group):
        return cls(group=group)

    def _get_plan(self, params):
        key = tuple(id(p) for _, p in params)
        if self._plan_key != key:
            device = params[0][1].device if params else self._device
            self._plan = ops.PackPlan([p for _, p in params], device)
            self._plan_key = key
            self._buf_a = DeviceBuffer(device=device)
            self._buf_b = DeviceBuffer(device=device)
        return self._plan

    def multi_node_mean_grad(self, model, zero_fill=False):
        params = self._grads_for(model, zero_fill)
        if not params:
            return
        plan = self._get_plan(params)
        wire_dtype = self.allreduce_grad_dtype or torch.float32
        flat = self._buf_a.array(plan.total, dtype=wire_dtype)
        ops.pack_grads(plan, flat)  # fused gather + cast (K1)
        dist.all_reduce(flat, group=self._group)
        # fused scatter + cast-back + ×1/size (K2)
        ops.unpack_grads(plan, flat, scale=1.0 / self._size)

    # -- async variant for comm/compute overlap ---------------------------
    def mean_grad_async(self, model, stream, zero_fill=False):
        """Pack + all-reduce on ``stream``; returns a handle.

        Call ``handle.unpack()`` (after making the consumer stream wait on
        ``handle.event``) to finish the mean-grad.  The flat buffer is
        communicator-owned so the torch allocator cannot recycle it while
        the collective is in flight.
        """
        params = self._grads_for(model, zero_fill)
        plan = self._get_plan(params)
        wire_dtype = self.allreduce_grad_dtype or torch.float32
        flat = self._buf_a.array(plan.total, dtype=wire_dtype)
        event = torch.cuda.Event()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            ops.pack_grads(plan, flat)
            dist.all_reduce(flat, group=self._group)
            event.record(stream)
        return _MeanGradHandle(self, plan, flat, event)


class _MeanGradHandle(object):

    def __init__(self, comm, plan, flat, event):
        self.comm = comm
        self.plan = plan
        self.flat = flat
        self.event = event

    def unpack(self):
        torch.cuda.current_stream().wait_event(self.event)
        ops.unpack_grads(self.plan, self.flat, scale=1.0 / self.comm.size)
