"""Communicator factory (reference: chainermn/communicators/__init__.py
create_communicator, ~L1-90).

Names keep the reference's vocabulary; ``pure_nccl`` is accepted as an
alias of ``pure_rccl`` (the collectives engine on ROCm is RCCL).  The
default is ``pure_rccl`` when a GPU is visible, else ``naive`` — mirroring
the late-Chainer default of pure_nccl.
"""

import torch

from chainermn_amd.communicators._base import CommunicatorBase  # noqa: F401
from chainermn_amd.communicators.dummy import DummyCommunicator
from chainermn_amd.communicators.flat import FlatCommunicator
from chainermn_amd.communicators.hierarchical import (
    HierarchicalCommunicator, NonCudaAwareCommunicator,
    SingleNodeCommunicator, TwoDimensionalCommunicator)
from chainermn_amd.communicators.naive import NaiveCommunicator
from chainermn_amd.communicators.pure_rccl import PureRcclCommunicator


def create_communicator(communicator_name=None, **kwargs):
    """Create a communicator by name.

    Args:
        communicator_name: one of ``pure_rccl`` (alias ``pure_nccl``),
            ``naive``, ``flat``, ``hierarchical``, ``two_dimensional``,
            ``single_node``, ``non_cuda_aware``, ``dummy``.  Default:
            ``pure_rccl`` with a GPU, ``naive`` without.
        **kwargs: forwarded to the communicator constructor
            (e.g. ``allreduce_grad_dtype='bf16'`` for pure_rccl).
    """
    if communicator_name is None:
        communicator_name = 'pure_rccl' if torch.cuda.is_available() else 'naive'
    table = {
        'pure_rccl': PureRcclCommunicator,
        'pure_nccl': PureRcclCommunicator,
        'naive': NaiveCommunicator,
        'flat': FlatCommunicator,
        'hierarchical': HierarchicalCommunicator,
        'two_dimensional': TwoDimensionalCommunicator,
        'single_node': SingleNodeCommunicator,
        'non_cuda_aware': NonCudaAwareCommunicator,
        'dummy': DummyCommunicator,
    }
    if communicator_name not in table:
        raise ValueError('unknown communicator: {} (choose from {})'.format(
            communicator_name, sorted(table)))
    cls = table[communicator_name]
    # reference-factory behaviour: tuning kwargs that a communicator does
    # not take (allreduce_grad_dtype / batched_copy on the non-pure_rccl
    # classes) are accepted and ignored with a warning
    import inspect
    accepted = set(inspect.signature(cls.__init__).parameters)
    dropped = [k for k in kwargs if k not in accepted]
    if dropped:
        import warnings
        warnings.warn('{} ignores kwargs {}'.format(
            communicator_name, dropped))
        kwargs = {k: v for k, v in kwargs.items() if k in accepted}
    return cls(**kwargs)
