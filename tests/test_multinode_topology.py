"""Multi-node topology emulation: 4 ranks pretend to be 2 nodes x 2 ranks
(hostname monkeypatch), exercising the intra/inter sub-group construction
and the staged reductions of hierarchical / two_dimensional /
non_cuda_aware exactly as a real 2-node job would drive them."""

import socket

import pytest
import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.testing import run_distributed


def _fake_two_nodes(rank):
    fake = "fakenode{}".format(rank // 2)
    socket.gethostname = lambda: fake


class ExampleModel(nn.Module):

    def __init__(self):
        super().__init__()
        self.a = nn.Linear(3, 4)
        self.b = nn.Linear(4, 2)


def _worker(rank, size, name):
    _fake_two_nodes(rank)
    comm = chainermn_amd.create_communicator(name)
    # geometry: 2 fake nodes x 2 ranks
    assert comm.intra_size == 2, (name, comm.intra_size)
    assert comm.inter_size == 2
    assert comm.intra_rank == rank % 2
    assert comm.inter_rank == rank // 2
    model = ExampleModel()
    for p in model.parameters():
        p.grad = torch.full_like(p.data, float(rank + 1))
    comm.multi_node_mean_grad(model)
    expect = sum(range(1, size + 1)) / size
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p.grad, expect)), \
            (name, float(p.grad.flatten()[0]), expect)
    return True


@pytest.mark.parametrize("name", ["hierarchical", "two_dimensional",
                                  "non_cuda_aware", "naive", "flat"])
def test_two_node_staged_reduction(name):
    assert run_distributed(_worker, 4, args=(name,)) == [True] * 4


def _single_node_rejects_worker(rank, size):
    _fake_two_nodes(rank)
    try:
        chainermn_amd.create_communicator("single_node")
        return False
    except ValueError:
        return True


def test_single_node_communicator_rejects_multinode():
    assert all(run_distributed(_single_node_rejects_worker, 4))
