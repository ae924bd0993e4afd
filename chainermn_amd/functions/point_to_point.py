"""Differentiable point-to-point communication.

Reference: chainermn/functions/point_to_point_communication.py (~L1-250):
``Send``'s forward ships activations and returns a *delegate variable* (an
empty variable holding the graph edge); its backward receives the upstream
gradient from the destination.  ``Recv`` mirrors it.  The delegate-variable
protocol sequences sends/recvs in the backward graph and prevents deadlock.

MI355X note: GPU payloads go over RCCL send/recv on xGMI via
torch.distributed; RCCL p2p has no tags, so ordering comes entirely from
the delegate discipline (issue order on the pair), which autograd's
reverse-topological replay preserves.
"""

import torch


class _Send(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, dest, tag, *inputs):
        ctx.comm = comm
        ctx.dest = dest
        ctx.tag = tag
        ctx.n_inputs = len(inputs)
        payload = inputs[0] if len(inputs) == 1 else tuple(inputs)
        comm.send(payload, dest, tag=tag)
        # The delegate variable: zero-dim, carries only the graph edge.
        device = inputs[0].device
        return torch.zeros((), device=device, dtype=inputs[0].dtype)

    @staticmethod
    def backward(ctx, grad_delegate):
        gx = ctx.comm.recv(ctx.dest, tag=ctx.tag,
                           force_tuple=ctx.n_inputs > 1)
        if ctx.n_inputs == 1:
            return (None, None, None, gx)
        return (None, None, None) + tuple(gx)


class _Recv(torch.autograd.Function):

    @staticmethod
    def forward(ctx, comm, source, tag, anchor):
        ctx.comm = comm
        ctx.source = source
        ctx.tag = tag
        ctx.anchor_shape = anchor.shape
        ctx.anchor_device = anchor.device
        ctx.anchor_dtype = anchor.dtype
        data = comm.recv(source, tag=tag, force_tuple=True)
        ctx.n_outputs = len(data)
        return data

    @staticmethod
    def backward(ctx, *grad_outputs):
        payload = (grad_outputs[0] if ctx.n_outputs == 1
                   else tuple(grad_outputs))
        ctx.comm.send(payload, ctx.source, tag=ctx.tag)
        g_anchor = torch.zeros(ctx.anchor_shape, device=ctx.anchor_device,
                               dtype=ctx.anchor_dtype)
        return (None, None, None, g_anchor)


def send(x, communicator, rank, tag=0):
    """Send ``x`` (tensor or tuple) to ``rank``; returns a delegate variable.

    The delegate must stay in this rank's graph (pass it to the next
    ``recv``'s ``delegate_variable=``, fold it through ``pseudo_connect``,
    or call ``.backward()`` on it) so the backward send/recv fires in order.
    """
    if communicator.rank == rank:
        raise ValueError('cannot send to myself (rank {})'.format(rank))
    inputs = (x,) if isinstance(x, torch.Tensor) else tuple(x)
    return _Send.apply(communicator, rank, tag, *inputs)


def recv(communicator, rank, delegate_variable=None, tag=0,
         force_tuple=False):
    """Receive tensors from ``rank``; differentiable (backward sends grad).

    ``delegate_variable`` — the delegate returned by an earlier ``send`` on
    this rank — orders this recv behind that send in the backward graph.
    """
    if communicator.rank == rank:
        raise ValueError('cannot receive from myself (rank {})'.format(rank))
    if delegate_variable is None:
        anchor = torch.zeros((), device=communicator.device,
                             requires_grad=True)
    else:
        anchor = delegate_variable
    out = _Recv.apply(communicator, rank, tag, anchor)
    if len(out) == 1 and not force_tuple:
        return out[0]
    return out
