"""Hierarchical / two-dimensional / single-node / non-CUDA-aware
communicators.

These reproduce the reference's multi-stage gradient reductions
(reference files: hierarchical_communicator.py ~L1-110,
two_dimensional_communicator.py ~L1-130, single_node_communicator.py
~L1-80, non_cuda_aware_communicator.py ~L1-150).  On a single 8×MI355X
node the intra stage IS the whole job, so hierarchical/two-dimensional
collapse to the flat all-reduce with extra structure kept for multi-node
parity (and exercised by the multi-process CPU tests, where "nodes" are
processes of one host — same emulation the reference's own test suite
used).
"""

import torch
import torch.distributed as dist

from chainermn_amd import ops
from chainermn_amd.communicators._base import CommunicatorBase
from chainermn_amd.communicators._node_groups import build_node_groups
from chainermn_amd.utils.memory import DeviceBuffer


class _FlatStagedCommunicator(CommunicatorBase):
    """Shared pack-flat → staged-reduce → unpack machinery."""

    def __init__(self, group=None):
        super().__init__(group=group)
        self._buffer = DeviceBuffer()
        self._plan = None
        self._plan_key = None
        (self._intra_group, self._intra_ranks,
         self._inter_group, self._node_lists) = build_node_groups(self)

    def _get_flat(self, params):
        key = tuple(id(p) for _, p in params)
        if self._plan_key != key:
            device = params[0][1].device if params else self._device
            self._plan = ops.PackPlan([p for _, p in params], device)
            self._plan_key = key
            self._buffer = DeviceBuffer(device=device)
        return self._plan, self._buffer.array(self._plan.total,
                                              dtype=torch.float32)

    def _local_root(self):
        return self._intra_ranks[0]

    def _staged_reduce(self, flat):
        raise NotImplementedError

    def multi_node_mean_grad(self, model, zero_fill=False):
        params = self._grads_for(model, zero_fill)
        if not params:
            return
        plan, flat = self._get_flat(params)
        ops.pack_grads(plan, flat)
        self._staged_reduce(flat)
        ops.unpack_grads(plan, flat, scale=1.0 / self._size)


class HierarchicalCommunicator(_FlatStagedCommunicator):
    """intra reduce → inter all-reduce among local roots → intra bcast."""

    def _staged_reduce(self, flat):
        root = self._local_root()
        dist.reduce(flat, dst=root, group=self._intra_group)
        if self._global_rank(self.rank) == root:
            dist.all_reduce(flat, group=self._inter_group)
        dist.broadcast(flat, src=root, group=self._intra_group)


class TwoDimensionalCommunicator(_FlatStagedCommunicator):
    """intra reduce-scatter → inter all-reduce of each shard → intra
    all-gather; keeps inter-node bytes at 1/intra_size."""

    def _staged_reduce(self, flat):
        n_local = len(self._intra_ranks)
        idx = self._intra_ranks.index(self._global_rank(self.rank))
        pad = (flat.numel() + n_local - 1) // n_local
        padded = flat
        if pad * n_local != flat.numel():
            padded = torch.zeros(pad * n_local, dtype=flat.dtype,
                                 device=flat.device)
            padded[:flat.numel()].copy_(flat)
        shards = list(padded.chunk(n_local))
        if flat.is_cuda:
            my_shard = torch.empty_like(shards[idx])
            dist.reduce_scatter(my_shard, shards, group=self._intra_group)
        else:
            # gloo has no reduce_scatter: per-shard reduce to its owner.
            for i, owner in enumerate(self._intra_ranks):
                dist.reduce(shards[i], dst=owner, group=self._intra_group)
            my_shard = shards[idx]
        dist.all_reduce(my_shard, group=self._inter_group)
        gathered = [torch.empty_like(my_shard) for _ in range(n_local)]
        dist.all_gather(gathered, my_shard, group=self._intra_group)
        packed = torch.cat(gathered)
        flat.copy_(packed[:flat.numel()])


class SingleNodeCommunicator(_FlatStagedCommunicator):
    """RCCL-only flat all-reduce; asserts the job fits one node.

    Closest ancestor of the 8×MI355X target config (reference:
    single_node_communicator.py).
    """

    def __init__(self, group=None):
        super().__init__(group=group)
        if self._size != self._intra_size:
            raise ValueError(
                'SingleNodeCommunicator requires all ranks on one node '
                '(size={} intra_size={})'.format(self._size, self._intra_size))

    def _staged_reduce(self, flat):
        dist.all_reduce(flat, group=self._group)


class NonCudaAwareCommunicator(_FlatStagedCommunicator):
    """intra reduce → D2H to pinned host → host (gloo) all-reduce among
    local roots → H2D → intra bcast.

    The reference needed this for MPI stacks that could not take device
    pointers; here the host stage runs over the gloo channel, which also
    makes this the one GPU communicator that works with no RCCL at all.
    """

    def __init__(self, group=None):
        super().__init__(group=group)
        self._host_buffer = DeviceBuffer(pinned=torch.cuda.is_available())

    def _staged_reduce(self, flat):
        root = self._local_root()
        dist.reduce(flat, dst=root, group=self._intra_group)
        if self._global_rank(self.rank) == root:
            host = self._host_buffer.array(flat.numel(), dtype=flat.dtype)
            host.copy_(flat)
            dist.all_reduce(host, group=self._inter_group)
            flat.copy_(host)
        dist.broadcast(flat, src=root, group=self._intra_group)
