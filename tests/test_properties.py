"""Property-based tests (hypothesis) for the layout-critical pure
functions: dataset splitting and flat-buffer planning."""

import numpy
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from chainermn_amd import ops
from chainermn_amd.datasets import _split


@settings(max_examples=200, deadline=None)
@given(n=st.integers(0, 500), size=st.integers(1, 9),
       force=st.booleans())
def test_split_covers_and_balances(n, size, force):
    order = numpy.arange(n)
    slices = _split(order, size, force)
    assert len(slices) == size
    covered = set()
    for s in slices:
        covered.update(int(i) for i in s)
    assert covered == set(range(n))
    lens = [len(s) for s in slices]
    if force and n >= size and n % size != 0:
        assert len(set(lens)) == 1  # equalised by wrap-padding
    elif n % size == 0:
        assert len(set(lens)) <= 1
    else:
        assert max(lens) - min(lens) <= 1


class _P:
    def __init__(self, n):
        self.data = torch.zeros(max(n, 1))
        self.grad = torch.zeros(max(n, 1))

    def numel(self):
        return self.data.numel()


@settings(max_examples=100, deadline=None)
@given(sizes=st.lists(st.integers(1, 5000), min_size=1, max_size=40))
def test_pack_plan_invariants(sizes):
    params = [_P(n) for n in sizes]
    plan = ops.PackPlan(params, torch.device('cpu'))
    # slots aligned, non-overlapping, inside the buffer
    prev_end = 0
    for off, n in zip(plan.offsets_cpu, plan.sizes_cpu):
        assert off % ops.ALIGN == 0
        assert off >= prev_end
        prev_end = off + n
    assert plan.total >= prev_end
    assert plan.total % ops.ALIGN == 0
    # pack/unpack on the CPU fallbacks round-trips exactly
    for p in params:
        p.grad = torch.randn_like(p.grad)
    flat = torch.zeros(plan.total)
    ops.pack_grads(plan, flat)
    originals = [p.grad.clone() for p in params]
    for p in params:
        p.grad.zero_()
    ops.unpack_grads(plan, flat, scale=1.0)
    for p, o in zip(params, originals):
        assert torch.equal(p.grad, o)
