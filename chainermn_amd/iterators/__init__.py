"""Multi-node iterators.

Reference: chainermn/iterators/multi_node_iterator.py (~L1-220):
``create_multi_node_iterator`` — the master rank iterates the real
iterator and broadcasts each batch (plus a stop flag and epoch state);
slaves replay.  ``create_synchronized_iterator`` broadcasts the RNG seed so
all ranks draw identical shuffles.  Used where every rank needs the same
batch (model parallelism).

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

    def __init__(self, actual_iterator, communicator, rank_master=0):
        self.actual = actual_iterator
        self.comm = communicator
        self.rank_master = rank_master
        self.epoch = 0
        self.is_new_epoch = False

    def __iter__(self):
        comm = self.comm
        if comm.rank == self.rank_master:
            for batch in self.actual:
                comm.bcast_obj(('batch', batch, self.epoch),
                               root=self.rank_master)
                yield batch
            self.epoch += 1
            comm.bcast_obj(('stop', None, self.epoch),
                           root=self.rank_master)
        else:
            while True:
                kind, batch, epoch = comm.bcast_obj(
                    None, root=self.rank_master)
                self.is_new_epoch = epoch != self.epoch
                self.epoch = epoch
                if kind == 'stop':
                    return
                yield batch

    def __len__(self):
        if hasattr(self.actual, '__len__'):
            return len(self.actual)
        raise TypeError('underlying iterator has no length')


def create_multi_node_iterator(actual_iterator, communicator, rank_master=0):
    return _MultiNodeIterator(actual_iterator, communicator, rank_master)


def create_synchronized_iterator(actual_iterator, communicator):
    """Synchronise torch RNG across ranks, then iterate locally: every rank
    draws identical shuffles (reference: create_synchronized_iterator)."""
    seed = communicator.bcast_obj(
        int(torch.initial_seed()) % (2 ** 63 - 1), root=0)
    torch.manual_seed(seed)
    return actual_iterator
