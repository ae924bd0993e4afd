"""Communicator suite — one parameterized run per communicator class.

Mirrors the reference's biggest test file
(tests/chainermn_tests/communicator_tests/test_communicator.py): an
ExampleModel gets known grads, multi_node_mean_grad must produce the exact
mean; plus bcast_data, send/recv (incl. tuples), every collective, the
*_obj ops and split.
"""

import pytest
import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.testing import run_distributed

COMM_NAMES = ['naive', 'flat', 'hierarchical', 'two_dimensional',
              'non_cuda_aware', 'single_node']


class ExampleModel(nn.Module):

    def __init__(self):
        super().__init__()
        self.a = nn.Linear(2, 3)
        self.b = nn.Linear(3, 4)
        self.c = nn.Linear(4, 5)


def _set_grads(model, value):
    for p in model.parameters():
        p.grad = torch.full_like(p.data, value)


def _mean_grad_worker(rank, size, name):
    comm = chainermn_amd.create_communicator(name)
    assert comm.rank == rank
    assert comm.size == size
    model = ExampleModel()
    _set_grads(model, float(rank + 1))
    comm.multi_node_mean_grad(model)
    expect = sum(range(1, size + 1)) / size
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p.grad, expect)), \
            '{}: {} != {}'.format(name, p.grad.flatten()[0], expect)
    # zero_fill: missing grads are treated as zeros
    model2 = ExampleModel()
    if rank == 0:
        _set_grads(model2, 2.0)
        model2.c.weight.grad = None
        model2.c.bias.grad = None
    else:
        _set_grads(model2, 0.0)
    comm.multi_node_mean_grad(model2, zero_fill=True)
    expect2 = 2.0 / size if rank is not None else 0
    assert torch.allclose(model2.a.weight.grad,
                          torch.full_like(model2.a.weight.grad, expect2))
    assert torch.allclose(model2.c.weight.grad,
                          torch.zeros_like(model2.c.weight.grad))
    return True


@pytest.mark.parametrize('name', COMM_NAMES)
def test_mean_grad(name):
    assert run_distributed(_mean_grad_worker, 2, args=(name,)) == [True, True]


def _bcast_data_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = ExampleModel()
    with torch.no_grad():
        for p in model.parameters():
            p.data.fill_(float(rank + 7))
    comm.bcast_data(model)
    for p in model.parameters():
        assert torch.allclose(p.data, torch.full_like(p.data, 7.0))
    return True


def test_bcast_data():
    run_distributed(_bcast_data_worker, 2)


def _p2p_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    # single tensor
    if rank == 0:
        comm.send(torch.arange(6, dtype=torch.float32).reshape(2, 3), 1)
    else:
        x = comm.recv(0)
        assert x.shape == (2, 3)
        assert torch.equal(x, torch.arange(6, dtype=torch.float32)
                           .reshape(2, 3))
    # tuple payload, reverse direction
    if rank == 1:
        comm.send((torch.ones(3), torch.zeros(2, 2, dtype=torch.float64)), 0)
    else:
        t = comm.recv(1)
        assert isinstance(t, tuple) and len(t) == 2
        assert t[1].dtype == torch.float64
    return True


def test_send_recv():
    run_distributed(_p2p_worker, 2)


def _collectives_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    # allreduce
    out = comm.allreduce(torch.full((3,), float(rank + 1)))
    assert torch.allclose(out, torch.full((3,), float(sum(range(1, size + 1)))))
    # bcast
    x = torch.arange(4, dtype=torch.float32) if rank == 0 else None
    out = comm.bcast(x, root=0)
    assert torch.equal(out, torch.arange(4, dtype=torch.float32))
    # allgather (different shapes per rank)
    xs = comm.allgather(torch.full((rank + 1,), float(rank)))
    assert len(xs) == size
    for r, t in enumerate(xs):
        assert t.shape == (r + 1,)
        assert torch.allclose(t, torch.full((r + 1,), float(r)))
    # alltoall
    outs = comm.alltoall([torch.full((2,), float(rank * 10 + j))
                          for j in range(size)])
    for src, t in enumerate(outs):
        assert torch.allclose(t, torch.full((2,), float(src * 10 + rank)))
    # gather
    g = comm.gather(torch.full((2,), float(rank)), root=1)
    if rank == 1:
        for r, t in enumerate(g):
            assert torch.allclose(t, torch.full((2,), float(r)))
    else:
        assert g is None
    # scatter
    if rank == 0:
        mine = comm.scatter([torch.full((2,), float(10 + r))
                             for r in range(size)], root=0)
    else:
        mine = comm.scatter(None, root=0)
    assert torch.allclose(mine, torch.full((2,), float(10 + rank)))
    return True


def test_collectives():
    run_distributed(_collectives_worker, 2)


def _obj_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    if rank == 0:
        comm.send_obj({'hello': [1, 2, 3]}, 1)
    else:
        assert comm.recv_obj(0) == {'hello': [1, 2, 3]}
    assert comm.bcast_obj('payload' if rank == 0 else None) == 'payload'
    g = comm.gather_obj(rank * 2, root=0)
    if rank == 0:
        assert g == [0, 2]
    assert comm.allgather_obj(rank) == list(range(size))
    assert comm.allreduce_obj(rank + 1) == sum(range(1, size + 1))
    assert comm.allreduce_obj({'v': float(rank)}) == \
        {'v': float(sum(range(size)))}
    return True


def test_obj_comms():
    run_distributed(_obj_worker, 2)


def _split_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    sub = comm.split(color=rank % 2, key=rank)
    assert sub.size == size // 2 + (1 if rank % 2 < size % 2 else 0)
    out = sub.allreduce(torch.tensor([float(rank)]))
    same_color = [r for r in range(size) if r % 2 == rank % 2]
    assert out.item() == float(sum(same_color))
    return True


def test_split():
    run_distributed(_split_worker, 4)


def _geometry_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    # all processes run on one host in this harness
    assert comm.intra_rank == rank
    assert comm.intra_size == size
    assert comm.inter_rank == 0
    assert comm.inter_size == 1
    return True


def test_geometry():
    run_distributed(_geometry_worker, 2)


def test_dummy_communicator():
    comm = chainermn_amd.create_communicator('dummy')
    assert comm.size == 1 and comm.rank == 0
    model = ExampleModel()
    _set_grads(model, 3.0)
    comm.multi_node_mean_grad(model)
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p.grad, 3.0))
    assert comm.allreduce_obj(5) == 5
    assert comm.bcast(torch.ones(2)) is not None


def test_unknown_name():
    with pytest.raises(ValueError):
        chainermn_amd.create_communicator('nonexistent')


def _chunked_obj_worker(rank, size):
    from chainermn_amd.communicators import _base
    _base.OBJ_CHUNK_BYTES = 1000  # force multi-chunk transport
    try:
        comm = chainermn_amd.create_communicator('naive')
        big = {'blob': bytes(range(256)) * 40, 'n': rank}  # ~10 KB
        if rank == 0:
            comm.send_obj(big, 1)
        else:
            got = comm.recv_obj(0)
            assert got['blob'] == big['blob'] and got['n'] == 0
        out = comm.bcast_obj(big if rank == 0 else None)
        assert out['blob'] == big['blob']
    finally:
        _base.OBJ_CHUNK_BYTES = 1 << 30
    return True


def test_chunked_obj_transport():
    run_distributed(_chunked_obj_worker, 2)


def _pure_rccl_cpu_worker(rank, size):
    """pure_rccl's pack->allreduce->unpack logic on the CPU fallbacks
    (fp32 wire; the GPU path is covered by the gpu-marked suite)."""
    comm = chainermn_amd.create_communicator('pure_rccl',
                                             allreduce_grad_dtype=None)
    model = ExampleModel()
    _set_grads(model, float(rank + 1))
    comm.multi_node_mean_grad(model)
    expect = sum(range(1, size + 1)) / size
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p.grad, expect))
    return True


def test_pure_rccl_cpu_logic():
    run_distributed(_pure_rccl_cpu_worker, 2)


def _split_of_split_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    half = comm.split(color=rank // 2, key=rank)    # {0,1} and {2,3}
    assert half.size == 2
    solo = half.split(color=half.rank, key=0)       # singletons
    assert solo.size == 1 and solo.rank == 0
    out = half.allreduce(torch.tensor([float(rank)]))
    mates = [r for r in range(size) if r // 2 == rank // 2]
    assert out.item() == float(sum(mates))
    return True


def test_split_of_split():
    run_distributed(_split_of_split_worker, 4)


def _pure_rccl_split_worker(rank, size):
    comm = chainermn_amd.create_communicator('pure_rccl',
                                             allreduce_grad_dtype=None)
    sub = comm.split(color=rank % 2, key=rank)
    assert type(sub).__name__ == 'PureRcclCommunicator'
    model = ExampleModel()
    _set_grads(model, float(rank + 1))
    sub.multi_node_mean_grad(model)
    mates = [r + 1 for r in range(size) if r % 2 == rank % 2]
    expect = sum(mates) / len(mates)
    for p in model.parameters():
        assert torch.allclose(p.grad, torch.full_like(p.grad, expect))
    return True


def test_pure_rccl_split_preserves_class():
    run_distributed(_pure_rccl_split_worker, 4)


def _world4_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    import torch.nn as nn
    torch.manual_seed(0)
    model = nn.Linear(10, 6)
    for i, p in enumerate(model.parameters()):
        p.grad = torch.full_like(p, float(rank + i))
    comm.multi_node_mean_grad(model)
    x = torch.full((8,), float(rank))
    s = comm.allreduce(x)
    return {'g0': next(model.parameters()).grad,
            's': s, 'rank': rank, 'size': comm.size,
            'split_color': comm.split(rank % 2, rank).rank}


def test_world4_mean_grad_and_split():
    """4-process world: mean-grad, allreduce and split (insurance for
    the driver's 8-GPU run — everything before it ran at world 2)."""
    results = run_distributed(_world4_worker, 4)
    mean0 = sum(range(4)) / 4.0
    for r in results:
        g = torch.from_numpy(r['g0'])
        assert torch.allclose(g, torch.full_like(g, mean0)), r['rank']
        assert torch.from_numpy(r['s']).allclose(
            torch.full((8,), 6.0))  # sum 0+1+2+3
        assert r['size'] == 4
        # split(color=rank%2): two sub-worlds of 2; sub-rank = index of
        # this rank among same-color ranks
        assert r['split_color'] == (r['rank'] // 2)
