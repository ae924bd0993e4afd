"""Multi-node iterators.

Reference: chainermn/iterators/multi_node_iterator.py (~L1-220):
``create_multi_node_iterator`` — the master rank iterates the real
iterator and broadcasts each batch (plus a stop flag and epoch state);
slaves replay.  ``create_synchronized_iterator`` synchronises the RNG
state across ranks so all ranks draw identical shuffles.  Used where
every rank needs the same batch (model parallelism).

Here an "iterator" is any iterable yielding torch tensors / tuples /
dicts / lists of them (e.g. a ``torch.utils.data.DataLoader``).
"""

import torch


def _tree_map(fn, obj):
    if isinstance(obj, torch.Tensor):
        return fn(obj)
    if isinstance(obj, dict):
        return {k: _tree_map(fn, v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_tree_map(fn, v) for v in obj)
    return obj


class _MultiNodeIterator(object):
    """Master broadcasts (kind, batch, epoch, is_new_epoch) per batch.

    The reference broadcast the iterator's epoch state with EVERY batch,
    so slave ranks track ``epoch``/``is_new_epoch`` in lockstep with the
    master — epoch-triggered extensions fire on the same iteration on
    every rank.  ``epoch`` counts completed passes; ``is_new_epoch`` is
    True exactly for the first batch of every pass after the first
    (i.e. the batch on which the previous epoch's wrap became visible).
    """

    def __init__(self, actual_iterator, communicator, rank_master=0):
        self.actual = actual_iterator
        self.comm = communicator
        self.rank_master = rank_master
        self.epoch = 0
        self.is_new_epoch = False

    def __iter__(self):
        comm = self.comm
        if comm.rank == self.rank_master:
            first = True
            for batch in self.actual:
                self.is_new_epoch = first and self.epoch > 0
                first = False
                comm.bcast_obj(
                    ('batch', batch, self.epoch, self.is_new_epoch),
                    root=self.rank_master)
                yield batch
            self.epoch += 1
            self.is_new_epoch = False
            comm.bcast_obj(('stop', None, self.epoch, False),
                           root=self.rank_master)
        else:
            while True:
                kind, batch, epoch, is_new = comm.bcast_obj(
                    None, root=self.rank_master)
                self.epoch = epoch
                self.is_new_epoch = is_new
                if kind == 'stop':
                    return
                yield batch

    def __len__(self):
        if hasattr(self.actual, '__len__'):
            return len(self.actual)
        raise TypeError('underlying iterator has no length')


def create_multi_node_iterator(actual_iterator, communicator, rank_master=0):
    return _MultiNodeIterator(actual_iterator, communicator, rank_master)


class _SynchronizedIterator(object):
    """Re-broadcasts rank 0's full torch RNG state before EVERY pass, so
    per-epoch shuffles stay identical even when ranks consume different
    amounts of randomness between epochs (the reference re-synced the RNG
    state per epoch, not just a seed at creation)."""

    def __init__(self, actual_iterator, communicator):
        self.actual = actual_iterator
        self.comm = communicator
        self._sync_rng()  # creation-time sync, like the reference

    def _sync_rng(self):
        state = None
        if self.comm.rank == 0:
            state = torch.get_rng_state()
        state = self.comm.bcast_obj(state, root=0)
        if not isinstance(state, torch.Tensor):
            state = torch.as_tensor(state)
        torch.set_rng_state(state.to(torch.uint8))

    def __iter__(self):
        self._sync_rng()
        return iter(self.actual)

    def __len__(self):
        return len(self.actual)


def create_synchronized_iterator(actual_iterator, communicator):
    """All ranks draw identical shuffles, every epoch (reference:
    create_synchronized_iterator)."""
    return _SynchronizedIterator(actual_iterator, communicator)
