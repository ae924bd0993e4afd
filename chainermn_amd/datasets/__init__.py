"""Dataset distribution.

Reference: chainermn/datasets/scatter_dataset.py (~L1-180) and
empty_dataset.py (~L1-30).  ``scatter_dataset`` pickles per-rank
sub-datasets at the root and ships them over the gloo object channel (the
reference's chunked MPI send_obj); ``scatter_index`` ships only index
ranges; ``create_empty_dataset`` builds length-preserving placeholders for
non-data (model-parallel follower) ranks.
"""

import numpy


def scatter_dataset(dataset, comm, root=0, shuffle=False, seed=None,
                    max_buf_len=256 * 1024 ** 2, force_equal_length=True):
    """Scatter ``dataset`` into near-equal per-rank shards.

    Root permutes indices (seeded), splits into ``comm.size`` slices (the
    first ``n % size`` ranks get one extra element; with
    ``force_equal_length`` shards wrap-pad to identical length so every
    rank runs the same number of iterations — keeping per-iteration
    collective counts rank-uniform).

    Note: items are pickled.  A torch tensor VIEW pickles its entire
    backing storage — materialize items (``.clone()``) or use
    ``scatter_index`` for large root-side datasets.
    """
    if comm.size == 1:
        return dataset
    if comm.rank == root:
        n = len(dataset)
        if shuffle:
            rng = numpy.random.RandomState(seed)
            order = rng.permutation(n)
        else:
            order = numpy.arange(n)
        slices = _split(order, comm.size, force_equal_length)
        shards = [[dataset[int(i)] for i in sl] for sl in slices]
        return comm.scatter_obj(shards, root=root, max_buf_len=max_buf_len)
    else:
        return comm.scatter_obj(None, root=root)


def scatter_index(n_total, comm, root=0, force_equal_length=True):
    """Scatter only index ranges (reference: scatter_index [v6+]).

    Returns this rank's list of dataset indices into a root-side dataset.
    """
    if comm.rank == root:
        order = numpy.arange(n_total)
        slices = _split(order, comm.size, force_equal_length)
        return comm.scatter_obj([s.tolist() for s in slices], root=root)
    return comm.scatter_obj(None, root=root)


def _split(order, size, force_equal_length):
    n = len(order)
    base = n // size
    rem = n % size
    slices = []
    off = 0
    for r in range(size):
        ln = base + (1 if r < rem else 0)
        slices.append(order[off:off + ln])
        off += ln
    if force_equal_length and rem != 0:
        target = base + 1
        slices = [numpy.concatenate([s, s[:target - len(s)]])
                  if len(s) < target else s for s in slices]
    return slices


class _EmptyDataset(object):

    def __init__(self, length):
        self._length = length

    def __len__(self):
        return self._length

    def __getitem__(self, i):
        if i < 0 or i >= self._length:
            raise IndexError(i)
        return ()


def create_empty_dataset(dataset):
    """Length-preserving dataset of empty tuples (reference:
    empty_dataset.py) so non-data ranks can drive an iterator/updater."""
    return _EmptyDataset(len(dataset))
