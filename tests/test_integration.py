"""End-to-end integration: the reference's CI smoke equivalents
(mpiexec -np 2 train_mnist.py --epoch 1), determinism across ranks, and
the global except hook's whole-job fail-fast."""

import os
import subprocess
import sys

import torch

import chainermn_amd
from chainermn_amd.testing import run_distributed

REPO = os.path.join(os.path.dirname(os.path.abspath(__file__)), '..')


def test_train_mnist_two_proc():
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29631',
         os.path.join(REPO, 'examples', 'mnist', 'train_mnist.py'),
         '--epoch', '1', '--batchsize', '256'],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-3000:]
    assert 'done:' in out.stdout


def _determinism_worker(rank, size):
    """Params must be BITWISE identical across ranks after DP steps
    (the rebuild's determinism check)."""
    import torch.nn as nn
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(0)
    model = nn.Linear(8, 8)
    opt = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.1), comm).setup(model)
    for step in range(3):
        opt.zero_grad(set_to_none=False)
        torch.manual_seed(77 + step * size + rank)
        model(torch.randn(4, 8)).pow(2).mean().backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_rankwise_bitwise_determinism():
    results = run_distributed(_determinism_worker, 2)
    import numpy
    for a, b in zip(results[0], results[1]):
        assert numpy.array_equal(a, b)  # bitwise, not allclose


def test_global_except_hook_kills_job():
    """One rank raising must fail the whole torchrun job promptly instead
    of deadlocking the collectives (reference: MPI_Abort semantics)."""
    script = r'''
import os, sys
sys.path.insert(0, %r)
import torch
import chainermn_amd
chainermn_amd.global_except_hook.add_hook()
comm = chainermn_amd.create_communicator('naive')
if comm.rank == 1:
    raise RuntimeError('boom on rank 1')
# rank 0 would block here forever without the hook killing the job
comm.allreduce(torch.ones(4))
''' % REPO
    import tempfile
    with tempfile.NamedTemporaryFile('w', suffix='.py',
                                     delete=False) as f:
        f.write(script)
        path = f.name
    try:
        out = subprocess.run(
            [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
             '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
             '--master-port', '29632', path],
            capture_output=True, text=True, timeout=120)
    finally:
        os.unlink(path)
    # the job must die promptly (no collective deadlock) with a nonzero
    # exit; torchrun routes the worker traceback to its log dir, so the
    # message itself may not surface in captured stderr
    assert out.returncode != 0
