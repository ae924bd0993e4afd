"""Differentiable communication functions: forward/backward numerical
checks across 2 ranks (reference: functions_tests/*)."""

import torch

import chainermn_amd
from chainermn_amd import functions
from chainermn_amd.testing import run_distributed


def _p2p_grad_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    # rank0: y = 3*x, send y to rank1.  rank1: z = recv^2, loss=z.sum().
    # d loss/dx on rank0 must be 2*(3x)*3 = 18x.
    if rank == 0:
        x = torch.arange(4, dtype=torch.float32, requires_grad=True)
        y = 3 * x
        delegate = functions.send(y, comm, 1)
        delegate.backward()
        assert torch.allclose(x.grad, 18 * torch.arange(4.0)), x.grad
    else:
        y = functions.recv(comm, 0)
        loss = (y * y).sum()
        loss.backward()
    return True


def test_p2p_gradients():
    run_distributed(_p2p_grad_worker, 2)


def _p2p_tuple_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    if rank == 0:
        a = torch.ones(3, requires_grad=True)
        b = torch.full((2,), 2.0, requires_grad=True)
        d = functions.send((a * 2, b * 3), comm, 1)
        d.backward()
        assert torch.allclose(a.grad, torch.full((3,), 2.0))
        assert torch.allclose(b.grad, torch.full((2,), 3.0))
    else:
        a, b = functions.recv(comm, 0, force_tuple=True)
        (a.sum() + b.sum()).backward()
    return True


def test_p2p_tuple_gradients():
    run_distributed(_p2p_tuple_worker, 2)


def _pipeline_roundtrip_worker(rank, size):
    """rank0 -> rank1 -> rank0 chain with pseudo_connect ordering."""
    comm = chainermn_amd.create_communicator('naive')
    if rank == 0:
        x = torch.ones(3, requires_grad=True) * 2
        x.retain_grad()
        d1 = functions.send(x, comm, 1)
        y = functions.recv(comm, 1, delegate_variable=d1)
        loss = y.sum()
        loss.backward()
        # chain: loss = sum(5 * x) => dx = 5
        assert torch.allclose(x.grad, torch.full((3,), 5.0)), x.grad
    else:
        h = functions.recv(comm, 0)
        out = functions.send(h * 5, comm, 0)
        out.backward()
    return True


def test_pipeline_roundtrip():
    run_distributed(_pipeline_roundtrip_worker, 2)


def _allgather_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    x = torch.full((2,), float(rank + 1), requires_grad=True)
    ys = functions.allgather(comm, x)
    assert torch.allclose(ys[0], torch.full((2,), 1.0))
    assert torch.allclose(ys[1], torch.full((2,), 2.0))
    # loss = sum_j (j+1) * ys[j]; dx_i = sum over ranks of (i+1) = size*(i+1)
    loss = sum((j + 1) * ys[j].sum() for j in range(size))
    loss.backward()
    assert torch.allclose(x.grad,
                          torch.full((2,), float(size * (rank + 1)))), x.grad
    return True


def test_allgather_grad():
    run_distributed(_allgather_worker, 2)


def _alltoall_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    xs = [torch.full((2,), float(rank * 10 + j), requires_grad=True)
          for j in range(size)]
    ys = functions.alltoall(comm, xs)
    for src in range(size):
        assert torch.allclose(ys[src],
                              torch.full((2,), float(src * 10 + rank)))
    loss = sum(float(rank + 1) * y.sum() for y in ys)
    loss.backward()
    # grad of xs[j] comes back from rank j with factor (j+1)
    for j in range(size):
        assert torch.allclose(xs[j].grad, torch.full((2,), float(j + 1)))
    return True


def test_alltoall_grad():
    run_distributed(_alltoall_worker, 2)


def _bcast_grad_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    if rank == 0:
        x = torch.full((3,), 2.0, requires_grad=True)
        y = functions.bcast(comm, x, root=0)
    else:
        y = functions.bcast(comm, None, root=0)
    assert torch.allclose(y, torch.full((3,), 2.0))
    ((rank + 1) * y.sum()).backward()
    if rank == 0:
        # grads summed over ranks: 1 + 2 = 3
        assert torch.allclose(x.grad, torch.full((3,), 3.0)), x.grad
    return True


def test_bcast_grad():
    run_distributed(_bcast_grad_worker, 2)


def _gather_scatter_grad_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    x = torch.full((2,), float(rank + 1), requires_grad=True)
    out = functions.gather(comm, x, root=0)
    if rank == 0:
        loss = sum((j + 2) * t.sum() for j, t in enumerate(out))
        loss.backward()
    else:
        out.backward()
    assert torch.allclose(x.grad, torch.full((2,), float(rank + 2))), x.grad

    if rank == 0:
        xs = [torch.full((2,), float(10 * (r + 1)), requires_grad=True)
              for r in range(size)]
        y = functions.scatter(comm, xs, root=0)
    else:
        y = functions.scatter(comm, None, root=0)
    assert torch.allclose(y, torch.full((2,), float(10 * (rank + 1))))
    ((rank + 3) * y.sum()).backward()
    if rank == 0:
        for r in range(size):
            assert torch.allclose(xs[r].grad,
                                  torch.full((2,), float(r + 3)))
    return True


def test_gather_scatter_grad():
    run_distributed(_gather_scatter_grad_worker, 2)


def _pseudo_connect_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    if rank == 0:
        x = torch.ones(2, requires_grad=True)
        d = functions.send(x * 4, comm, 1)
        z = torch.full((2,), 3.0, requires_grad=True)
        tied = functions.pseudo_connect(d, z)
        (tied.sum() * 2).backward()
        assert torch.allclose(z.grad, torch.full((2,), 2.0))
        # send backward also ran (received grad 1.0 from rank 1) => 4.0
        assert torch.allclose(x.grad, torch.full((2,), 4.0))
    else:
        y = functions.recv(comm, 0)
        y.sum().backward()
    return True


def test_pseudo_connect():
    run_distributed(_pseudo_connect_worker, 2)


def _allreduce_fn_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    x = torch.full((3,), float(rank + 1), requires_grad=True)
    y = functions.allreduce(comm, x)
    assert torch.allclose(y, torch.full((3,), 3.0))  # 1 + 2
    ((rank + 1.0) * y.sum()).backward()
    # dy/dx gets all-reduced too: sum of per-rank gy = 1 + 2 = 3
    assert torch.allclose(x.grad, torch.full((3,), 3.0)), x.grad
    return True


def test_allreduce_function_grad():
    run_distributed(_allreduce_fn_worker, 2)
