"""multi_node_snapshot — replica-set-deduplicated snapshots.

Reference: chainermn/extensions/multi_node_snapshot.py [v7] (~L1-170):
wraps a snapshot extension; ranks are partitioned into replica sets (ranks
holding identical — data-parallel — state); only each set's leader writes,
others no-op.  Distinct state per model-parallel shard still gets written
(one file per replica set).
"""

import os

import torch


class _SnapshotExtension(object):
    """Default snapshot writer: trainer state to {out}/{filename}."""

    name = 'snapshot'

    def __init__(self, filename='snapshot_iter_{iteration}'):
        self.filename = filename

    def __call__(self, trainer):
        path = os.path.join(
            trainer.out, self.filename.format(iteration=trainer.iteration))
        torch.save(trainer.state_dict(), path)


def snapshot(filename='snapshot_iter_{iteration}'):
    return _SnapshotExtension(filename)


def multi_node_snapshot(comm, snapshot_ext, replica_sets):
    """Wrap ``snapshot_ext`` so only replica-set leaders write.

    ``replica_sets``: iterable of rank lists; ranks not mentioned form
    singleton sets.  E.g. 2-way model parallel × 4-way data parallel on 8
    ranks: ``[[0, 2, 4, 6], [1, 3, 5, 7]]`` → ranks 0 and 1 write.
    """
    sets = [list(s) for s in replica_sets]
    mentioned = {r for s in sets for r in s}
    for r in range(comm.size):
        if r not in mentioned:
            sets.append([r])
    my_set = None
    for s in sets:
        if comm.rank in s:
            my_set = s
            break
    leader = min(my_set)

    class _MultiNodeSnapshot(object):
        name = 'MultiNodeSnapshot'

        def __call__(self, trainer):
            if comm.rank == leader:
                snapshot_ext(trainer)

    return _MultiNodeSnapshot()
