"""Data-parallel optimizer wrappers.

Reference: chainermn/optimizers.py (~L1-300): ``create_multi_node_optimizer``
wraps a Chainer optimizer so every ``update()`` first averages gradients
across ranks; ``double_buffering=True`` overlaps the all-reduce of step n
with the forward/backward of step n+1 using a worker thread + dedicated
stream (1-step-stale gradients).

Torch/MI355X redesign: the wrapper proxies a ``torch.optim.Optimizer``;
double buffering needs no thread — the pack+all-reduce is enqueued on a
side HIP stream and fenced with events (HIP stream semantics make the
reference's Python worker thread pure overhead).  Gradients live in one
flat fp32 arena (every ``param.grad`` is a view), so the "pack" for a
bf16/fp16 wire format is a single fused cast kernel and the arena is
framework-owned — the reference's DeviceMemory lifetime trick
(grads must outlive the async all-reduce) holds by construction.
"""

import torch
import torch.distributed as dist

from chainermn_amd import ops
from chainermn_amd.communicators.pure_rccl import PureRcclCommunicator


def create_multi_node_optimizer(actual_optimizer, communicator,
                                double_buffering=False, zero_fill=False):
    """Wrap a torch.optim.Optimizer for data-parallel training.

    Usage::

        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        opt = chainermn_amd.create_multi_node_optimizer(opt, comm)
        opt.setup(model)
        ...
        loss.backward(); opt.step(); opt.zero_grad()
    """
    if double_buffering:
        if not isinstance(communicator, PureRcclCommunicator):
            raise ValueError(
                'double_buffering requires a PureRcclCommunicator')
        return DoubleBufferingOptimizer(actual_optimizer, communicator,
                                        zero_fill)
    return _MultiNodeOptimizer(actual_optimizer, communicator, zero_fill)


class _MultiNodeOptimizer(object):

    def __init__(self, actual_optimizer, communicator, zero_fill=False):
        # __dict__ poking keeps __getattr__/__setattr__ proxying clean
        # (same trick as the reference's _MultiNodeOptimizer).
        super().__setattr__('communicator', communicator)
        super().__setattr__('actual_optimizer', actual_optimizer)
        super().__setattr__('target', None)
        super().__setattr__('zero_fill', zero_fill)

    def setup(self, model, broadcast=True):
        """Bind the model whose grads are averaged; broadcast initial params
        from rank 0 so every rank starts identical."""
        super().__setattr__('target', model)
        if broadcast:
            self.communicator.bcast_data(model)
        return self

    def step(self, closure=None):
        if self.target is None:
            raise RuntimeError('call setup(model) before step()')
        loss = None
        if closure is not None:
            loss = closure()
        self.communicator.multi_node_mean_grad(self.target, self.zero_fill)
        self.actual_optimizer.step()
        return loss

    # chainermn name: update()
    def update(self, lossfun=None, *args, **kwargs):
        if lossfun is not None:
            loss = lossfun(*args, **kwargs)
            self.actual_optimizer.zero_grad(set_to_none=False)
            loss.backward()
            self.step()
            return loss
        return self.step()

    def __getattr__(self, name):
        return getattr(self.actual_optimizer, name)

    def __setattr__(self, name, value):
        if name in self.__dict__:
            super().__setattr__(name, value)
        else:
            setattr(self.actual_optimizer, name, value)


class GradArena(object):
    """One flat fp32 gradient arena; every param.grad is a view into it.

    Replaces the reference's per-step pack (K1): gradients accumulate
    directly at their flat offsets during backward, so the wire-format cast
    is one fused kernel over the arena and the fp32 all-reduce needs no
    copy at all.
    """

    def __init__(self, model):
        params = [p for _, p in sorted(model.named_parameters())
                  if p.requires_grad]
        self.plan = ops.PackPlan(params, params[0].device)
        self.flat = torch.zeros(self.plan.total, dtype=torch.float32,
                                device=params[0].device)
        for p, off, n in zip(params, self.plan.offsets_cpu,
                             self.plan.sizes_cpu):
            p.grad = self.flat[off:off + n].view_as(p.data)
        self.params = params

    def zero_(self):
        self.flat.zero_()


class DoubleBufferingOptimizer(object):
    """Overlapped data-parallel optimizer (1-step-stale gradients).

    Step *n*: launch cast+all-reduce of grads *n* on a side HIP stream,
    then apply the update with the all-reduced grads of step *n-1*
    (reference semantics: optimizers.py _DoubleBufferingOptimizer
    ~L120-300).  The first step performs no parameter update.
    """

    def __init__(self, actual_optimizer, communicator, zero_fill=False):
        super().__setattr__('communicator', communicator)
        super().__setattr__('actual_optimizer', actual_optimizer)
        super().__setattr__('target', None)
        super().__setattr__('zero_fill', zero_fill)
        super().__setattr__('_state', None)

    def setup(self, model, broadcast=True):
        super().__setattr__('target', model)
        if broadcast:
            self.communicator.bcast_data(model)
        arena = GradArena(model)
        wire = self.communicator.allreduce_grad_dtype or torch.float32
        dev = arena.flat.device
        state = {
            'arena': arena,
            'wire': [torch.zeros(arena.plan.total, dtype=wire, device=dev)
                     for _ in range(2)],
            'ev_pack': torch.cuda.Event(),
            'ev_ar': [torch.cuda.Event(), torch.cuda.Event()],
            'stream': torch.cuda.Stream(),
            'cur': 0,
            'step': 0,
        }
        super().__setattr__('_state', state)
        return self

    def step(self, closure=None):
        if self._state is None:
            raise RuntimeError('call setup(model) before step()')
        loss = closure() if closure is not None else None
        s = self._state
        cur, prev = s['cur'], 1 - s['cur']
        arena, stream = s['arena'], s['stream']
        main = torch.cuda.current_stream()
        stream.wait_stream(main)
        with torch.cuda.stream(stream):
            # fused pack = cast fp32 arena -> wire buffer (no gather needed)
            ops.cast_flat(arena.flat, s['wire'][cur])
            s['ev_pack'].record(stream)
            dist.all_reduce(s['wire'][cur], group=self.communicator.group)
            s['ev_ar'][cur].record(stream)
        main.wait_event(s['ev_pack'])   # arena free to overwrite
        if s['step'] > 0:
            main.wait_event(s['ev_ar'][prev])
            # fused unpack: cast-back + x(1/size) into the arena the
            # wrapped optimizer reads through param.grad views.
            ops.cast_flat(s['wire'][prev], arena.flat,
                          scale=1.0 / self.communicator.size)
            self.actual_optimizer.step()
        s['step'] += 1
        s['cur'] = prev
        return loss

    # chainermn name: update()
    def update(self, lossfun=None, *args, **kwargs):
        if lossfun is not None:
            loss = lossfun(*args, **kwargs)
            self.zero_grad()
            loss.backward()
            self.step()
            return loss
        return self.step()

    def zero_grad(self, set_to_none=False):
        # Views must stay bound to the arena: never set_to_none.
        self._state['arena'].zero_()

    def wait(self):
        """Drain in-flight communication (call before checkpoint/eval)."""
        torch.cuda.current_stream().wait_event(
            self._state['ev_ar'][1 - self._state['cur']])
        torch.cuda.synchronize()

    def __getattr__(self, name):
        return getattr(self.actual_optimizer, name)

    def __setattr__(self, name, value):
        if name in self.__dict__:
            super().__setattr__(name, value)
        else:
            setattr(self.actual_optimizer, name, value)
