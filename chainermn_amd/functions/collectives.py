"""Differentiable collective communication.

Reference: chainermn/functions/collective_communication.py (~L1-260):
AllGather (bwd = all-to-all + local sum is reduce-scatter-equivalent),
AllToAll (bwd = all-to-all), Bcast (bwd = reduce to root), Gather
(bwd = scatter), Scatter (bwd = gather).  Payloads are RCCL collectives
over xGMI on GPU, gloo on CPU.
"""

import torch


class _AllGather(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, x):
        ctx.comm = comm
        return comm.allgather(x)

    @staticmethod
    def backward(ctx, *grad_outputs):
        comm = ctx.comm
        scattered = comm.alltoall([g.contiguous() for g in grad_outputs])
        gx = scattered[0].clone()
        for g in scattered[1:]:
            gx += g
        return None, gx


class _AllToAll(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, *xs):
        ctx.comm = comm
        return ctx.comm.alltoall(xs)

    @staticmethod
    def backward(ctx, *grad_outputs):
        gxs = ctx.comm.alltoall([g.contiguous() for g in grad_outputs])
        return (None,) + tuple(gxs)


class _Bcast(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, root, x):
        ctx.comm = comm
        ctx.root = root
        return comm.bcast(x, root=root)

    @staticmethod
    def backward(ctx, grad_output):
        gsum = ctx.comm.allreduce(grad_output)
        if ctx.comm.rank == ctx.root:
            return None, None, gsum
        return None, None, None


class _Gather(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, root, x):
        ctx.comm = comm
        ctx.root = root
        out = comm.gather(x, root=root)
        if comm.rank == root:
            return out
        # Non-root ranks get a delegate so backward fires (grad arrives
        # via the root's scatter).
        return torch.zeros((), device=x.device, dtype=x.dtype)

    @staticmethod
    def backward(ctx, *grad_outputs):
        comm = ctx.comm
        if comm.rank == ctx.root:
            gx = comm.scatter([g.contiguous() for g in grad_outputs],
                              root=ctx.root)
        else:
            gx = comm.scatter(None, root=ctx.root)
        return None, None, gx


class _Scatter(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, root, *xs):
        ctx.comm = comm
        ctx.root = root
        ctx.n_inputs = len(xs)
        if comm.rank == root:
            return comm.scatter(xs, root=root)
        return comm.scatter(None, root=root)

    @staticmethod
    def backward(ctx, grad_output):
        comm = ctx.comm
        gathered = comm.gather(grad_output.contiguous(), root=ctx.root)
        if comm.rank == ctx.root:
            return (None, None) + tuple(gathered)
        return (None, None) + (None,) * ctx.n_inputs


class _AllReduce(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, x):
        ctx.comm = comm
        return comm.allreduce(x)

    @staticmethod
    def backward(ctx, grad_output):
        # d(sum_r x_r)/dx_this receives gy from every rank: all-reduce.
        return None, ctx.comm.allreduce(grad_output.contiguous())


def allgather(communicator, x):
    """Differentiable all-gather; returns a tuple of tensors."""
    return _AllGather.apply(communicator, x)


def alltoall(communicator, xs):
    """Differentiable all-to-all over a sequence of ``size`` tensors."""
    if len(xs) != communicator.size:
        raise ValueError('alltoall requires len(xs) == size')
    return _AllToAll.apply(communicator, *xs)


def bcast(communicator, x, root=0):
    """Differentiable broadcast (non-root may pass None)."""
    if communicator.rank != root and x is None:
        x = torch.zeros((), device=communicator.device, requires_grad=True)
    return _Bcast.apply(communicator, root, x)


def gather(communicator, x, root=0):
    """Differentiable gather to root (tuple at root, delegate elsewhere)."""
    return _Gather.apply(communicator, root, x)


def scatter(communicator, xs, root=0):
    """Differentiable scatter from root; returns this rank's tensor."""
    if communicator.rank == root:
        return _Scatter.apply(communicator, root, *xs)
    anchor = torch.zeros((), device=communicator.device, requires_grad=True)
    return _Scatter.apply(communicator, root, anchor)


def allreduce(communicator, x):
    """Differentiable all-reduce (sum)."""
    return _AllReduce.apply(communicator, x)
