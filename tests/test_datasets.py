"""Dataset scatter / iterators (reference: datasets_tests +
iterators_tests): union of shards covers the dataset, balance, equal
length; multi-node iterator replays identical batches."""

import torch

import chainermn_amd
from chainermn_amd.datasets import (create_empty_dataset, scatter_dataset,
                                    scatter_index)
from chainermn_amd.iterators import (create_multi_node_iterator,
                                     create_synchronized_iterator)
from chainermn_amd.testing import run_distributed


def _scatter_worker(rank, size, n, shuffle, force_equal):
    comm = chainermn_amd.create_communicator('naive')
    dataset = list(range(n))
    shard = scatter_dataset(dataset, comm, shuffle=shuffle, seed=42,
                            force_equal_length=force_equal)
    return list(shard)


def test_scatter_dataset_covers_and_balances():
    for n in (10, 11):
        shards = run_distributed(_scatter_worker, 2, args=(n, False, False))
        union = sorted(shards[0] + shards[1])
        assert union == list(range(n))
        assert abs(len(shards[0]) - len(shards[1])) <= 1


def test_scatter_dataset_equal_length_pads():
    shards = run_distributed(_scatter_worker, 2, args=(11, False, True))
    assert len(shards[0]) == len(shards[1]) == 6
    covered = set(shards[0]) | set(shards[1])
    assert covered == set(range(11))


def test_scatter_dataset_shuffle_seeded():
    a = run_distributed(_scatter_worker, 2, args=(10, True, False))
    b = run_distributed(_scatter_worker, 2, args=(10, True, False))
    assert a == b  # same seed -> same permutation
    assert sorted(a[0] + a[1]) == list(range(10))


def _scatter_index_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    idx = scatter_index(10, comm)
    return idx


def test_scatter_index():
    shards = run_distributed(_scatter_index_worker, 2)
    assert sorted(shards[0] + shards[1]) == list(range(10))


def test_create_empty_dataset():
    ds = create_empty_dataset([1, 2, 3])
    assert len(ds) == 3
    assert ds[0] == ()
    try:
        ds[3]
        assert False
    except IndexError:
        pass


def _mn_iterator_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    base = [torch.tensor([i, i + 1]) for i in range(4)] if rank == 0 else None
    it = create_multi_node_iterator(base, comm, rank_master=0)
    seen = [b.tolist() for b in it]
    return seen


def test_multi_node_iterator_replays():
    results = run_distributed(_mn_iterator_worker, 2)
    assert results[0] == results[1]
    assert results[0] == [[0, 1], [1, 2], [2, 3], [3, 4]]


def _sync_iterator_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(100 + rank)  # deliberately different
    it = create_synchronized_iterator(range(3), comm)
    draws = torch.randperm(8).tolist()
    return (list(it), draws)


def test_synchronized_iterator_syncs_rng():
    results = run_distributed(_sync_iterator_worker, 2)
    assert results[0][0] == results[1][0] == [0, 1, 2]
    assert results[0][1] == results[1][1]


def _mn_iterator_epochs_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    base = [torch.tensor([i]) for i in range(3)] if rank == 0 else None
    it = create_multi_node_iterator(base, comm, rank_master=0)
    seen = []
    for epoch in range(2):
        for b in it:
            seen.append((epoch, int(b[0])))
    return seen


def test_multi_node_iterator_multiple_epochs():
    results = run_distributed(_mn_iterator_epochs_worker, 2)
    assert results[0] == results[1]
    assert results[0] == [(0, 0), (0, 1), (0, 2), (1, 0), (1, 1), (1, 2)]


def _mn_iterator_epoch_state_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    base = [torch.tensor([i]) for i in range(3)] if rank == 0 else None
    it = create_multi_node_iterator(base, comm, rank_master=0)
    trace = []
    for _ in range(2):
        for b in it:
            trace.append((it.epoch, it.is_new_epoch, int(b[0])))
        trace.append(('end', it.epoch))
    return trace


def test_multi_node_iterator_epoch_state_on_slaves():
    """Slaves see per-batch epoch/is_new_epoch identical to the master
    (epoch-triggered extensions fire on the same iteration everywhere)."""
    results = run_distributed(_mn_iterator_epoch_state_worker, 2)
    assert results[0] == results[1]
    assert results[0] == [
        (0, False, 0), (0, False, 1), (0, False, 2), ('end', 1),
        (1, True, 0), (1, False, 1), (1, False, 2), ('end', 2)]


def _sync_iterator_reseed_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(100 + rank)  # deliberately different

    class Shuffler:
        def __iter__(self):
            return iter(torch.randperm(16).tolist())

        def __len__(self):
            return 16

    it = create_synchronized_iterator(Shuffler(), comm)
    epochs = []
    for _ in range(3):
        epochs.append(list(it))
        # consume a DIFFERENT amount of randomness per rank between
        # epochs — a one-time seed sync would diverge here
        for _ in range(rank + 1):
            torch.rand(5)
    return epochs


def test_synchronized_iterator_resyncs_every_epoch():
    results = run_distributed(_sync_iterator_reseed_worker, 2)
    assert results[0] == results[1]
    assert len(results[0]) == 3
    # shuffles differ across epochs (the RNG advances on rank 0)
    assert results[0][0] != results[0][1]
