"""pseudo_connect — graph glue for the delegate-variable protocol.

Reference: chainermn/functions/pseudo_connect.py (~L1-100): forwards the
actual variables unchanged while recording a graph edge to the delegate
variable, so ``loss.backward()`` reaches earlier pipeline stages' ``Send``
nodes in the right order.
"""

import torch


class _PseudoConnect(torch.autograd.Function):

    @staticmethod
    def forward(ctx, delegate, *actuals):
        ctx.delegate_meta = (delegate.shape, delegate.device, delegate.dtype)
        # view_as creates fresh graph nodes without copying storage.
        return tuple(a.view_as(a) for a in actuals)

    @staticmethod
    def backward(ctx, *grad_outputs):
        shape, device, dtype = ctx.delegate_meta
        g_delegate = torch.zeros(shape, device=device, dtype=dtype)
        return (g_delegate,) + tuple(grad_outputs)


def pseudo_connect(delegate_variable, *actual_variables):
    """Ties ``actual_variables`` to ``delegate_variable`` in the graph."""
    if delegate_variable is None:
        raise ValueError('delegate_variable must not be None')
    out = _PseudoConnect.apply(delegate_variable, *actual_variables)
    if len(out) == 1:
        return out[0]
    return out
