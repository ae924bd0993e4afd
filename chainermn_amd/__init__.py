"""chainermn_amd — MI355X-native distributed training framework.

A from-scratch rebuild of ChainerMN's capabilities (reference:
chainer/chainermn v1.3.1) designed MI355X-first:

* PyTorch-ROCm is the host framework (``torch.nn.Module`` replaces
  ``chainer.Link``, ``torch.autograd.Function`` replaces ``FunctionNode``).
* Collectives run over RCCL (``torch.distributed`` backend ``"nccl"`` is RCCL
  on ROCm) across xGMI; the object/control plane runs over a gloo CPU channel
  (the reference used MPI — SURVEY.md §2.3).
* Gradient-path device work (pack/unpack/cast, fused optimizer steps,
  multi-node BN statistics) is hand-written CDNA4 HIP (gfx950) in
  ``chainermn_amd/_hip/`` — the reference delegated these to CuPy JIT kernels
  (SURVEY.md §2.4).

Public API mirrors ``chainermn``'s package facade
(reference: chainermn/__init__.py:~L1-40).
"""

from chainermn_amd.communicators import CommunicatorBase  # noqa: F401
from chainermn_amd.communicators import create_communicator  # noqa: F401
from chainermn_amd.optimizers import create_multi_node_optimizer  # noqa: F401
from chainermn_amd.optimizers import DoubleBufferingOptimizer  # noqa: F401
from chainermn_amd.evaluators import create_multi_node_evaluator  # noqa: F401
from chainermn_amd.datasets import scatter_dataset  # noqa: F401
from chainermn_amd.datasets import scatter_index  # noqa: F401
from chainermn_amd.datasets import create_empty_dataset  # noqa: F401
from chainermn_amd import global_except_hook  # noqa: F401
from chainermn_amd import functions  # noqa: F401
from chainermn_amd import links  # noqa: F401
from chainermn_amd import extensions  # noqa: F401
from chainermn_amd import iterators  # noqa: F401

__version__ = '0.1.0'
