"""Trainer extensions (reference: extensions_tests): checkpointer
save→kill→maybe_load→state equality + GC window; evaluator averaging;
AllreducePersistent; ObservationAggregator; multi_node_snapshot."""

import os
import tempfile

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.extensions import (AllreducePersistent,
                                      ObservationAggregator,
                                      create_multi_node_checkpointer,
                                      multi_node_snapshot)
from chainermn_amd.extensions.multi_node_snapshot import snapshot
from chainermn_amd.testing import run_distributed
from chainermn_amd.training import Evaluator, Trainer


def _make_trainer(model, comm, n_iters=10, out='result'):
    opt = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.01), comm).setup(model)
    data = [(torch.randn(4, 3), ) for _ in range(5)]

    def loss_fn(model, batch):
        return model(batch[0]).pow(2).mean()

    return Trainer(model, opt, data, (n_iters, 'iteration'),
                   loss_fn=loss_fn, out=out)


def _checkpoint_worker(rank, size, tmpdir):
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(0)
    model = nn.Linear(3, 2)
    trainer = _make_trainer(model, comm, n_iters=6, out=tmpdir)
    cp = create_multi_node_checkpointer('ck', comm, cp_interval=2,
                                        gc_interval=3, path=tmpdir)
    trainer.extend(cp, trigger=(2, 'iteration'))
    trainer.run()
    saved_w = model.weight.detach().clone()

    # "restart": fresh trainer + maybe_load
    torch.manual_seed(99)
    model2 = nn.Linear(3, 2)
    trainer2 = _make_trainer(model2, comm, n_iters=6, out=tmpdir)
    cp2 = create_multi_node_checkpointer('ck', comm, cp_interval=2,
                                         gc_interval=3, path=tmpdir)
    it = cp2.maybe_load(trainer2)
    assert it == 6, it
    assert trainer2.iteration == 6
    assert torch.allclose(model2.weight, saved_w)
    stats = cp.get_stats()
    if rank == 0:
        assert stats['count'] > 0
    # GC window: at most gc_interval iterations on disk per rank
    files = [f for f in os.listdir(tmpdir)
             if f.startswith('ck.') and f.endswith('.' + str(rank))]
    assert len(files) <= 3, files
    return True


def test_checkpointer_roundtrip():
    with tempfile.TemporaryDirectory() as tmpdir:
        run_distributed(_checkpoint_worker, 2, args=(tmpdir,))


def _evaluator_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.Linear(3, 2)
    with torch.no_grad():
        model.weight.fill_(0.0)
        model.bias.fill_(float(rank + 1))
    data = [(torch.zeros(2, 3),)]

    def loss_fn(model, batch):
        return model(batch[0]).sum()

    ev = Evaluator(data, model, loss_fn=loss_fn)
    ev = chainermn_amd.create_multi_node_evaluator(ev, comm)
    result = ev.evaluate()
    # local losses: rank0: 2*(1+1)=4, rank1: 2*(2+2)=8 -> mean 6
    assert abs(result['validation/main/loss'] - 6.0) < 1e-6, result
    return True


def test_multi_node_evaluator():
    run_distributed(_evaluator_worker, 2)


def _allreduce_persistent_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.BatchNorm1d(3)
    with torch.no_grad():
        model.running_mean.fill_(float(rank))
    AllreducePersistent(model, comm)()
    assert torch.allclose(model.running_mean,
                          torch.full((3,), 0.5))
    # integer buffers (num_batches_tracked) untouched
    assert model.num_batches_tracked.item() == 0
    return True


def test_allreduce_persistent():
    run_distributed(_allreduce_persistent_worker, 2)


def _aggregator_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.Linear(3, 2)
    trainer = _make_trainer(model, comm, n_iters=2)
    agg = ObservationAggregator(comm, 'main/loss', 'main/loss_avg')
    trainer.extend(agg, trigger=(1, 'iteration'))
    trainer.run()
    last = trainer.observations_history[-1]
    return {'loss': last['main/loss'], 'avg': last['main/loss_avg']}


def test_observation_aggregator():
    results = run_distributed(_aggregator_worker, 2)
    expect = (results[0]['loss'] + results[1]['loss']) / 2
    assert abs(results[0]['avg'] - expect) < 1e-9
    assert results[0]['avg'] == results[1]['avg']


def _snapshot_worker(rank, size, tmpdir):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.Linear(3, 2)
    trainer = _make_trainer(model, comm, n_iters=2, out=tmpdir)
    snap = multi_node_snapshot(comm, snapshot('snap_{iteration}.r%d' % rank),
                               replica_sets=[[0, 1]])
    trainer.extend(snap, trigger=(2, 'iteration'))
    trainer.run()
    return sorted(os.listdir(tmpdir))


def test_multi_node_snapshot_dedup():
    with tempfile.TemporaryDirectory() as tmpdir:
        results = run_distributed(_snapshot_worker, 2, args=(tmpdir,))
    # only the replica-set leader (rank 0) wrote its file
    files = [f for f in results[0] if f.startswith('snap_')]
    assert files == ['snap_2.r0'], results[0]


def _evaluator_metrics_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    model = nn.Linear(3, 2)
    with torch.no_grad():
        model.weight.zero_()
        model.bias.fill_(float(rank))
    data = [(torch.zeros(4, 3),)]

    def loss_fn(model, batch):
        return model(batch[0]).sum()

    def acc_metric(model, batch):
        return float(rank + 1)  # rank0: 1, rank1: 2 -> mean 1.5

    ev = chainermn_amd.create_multi_node_evaluator(
        Evaluator(data, model, loss_fn=loss_fn,
                  metrics={'accuracy': acc_metric}), comm)
    r = ev.evaluate()
    assert abs(r['validation/main/accuracy'] - 1.5) < 1e-9
    return True


def test_evaluator_metrics_averaged():
    run_distributed(_evaluator_metrics_worker, 2)
