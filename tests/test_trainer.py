"""Trainer protocol: triggers, priorities, state, reports (the host-side
surface the reference's extensions plugged into)."""

import json
import os
import tempfile

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.training import LogReport, PrintReport, Trainer


def _trainer(n_iters=6, out=None):
    comm = chainermn_amd.create_communicator('dummy')
    model = nn.Linear(3, 2)
    opt = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=0.01), comm).setup(model)
    data = [(torch.randn(4, 3),) for _ in range(3)]

    def loss_fn(model, batch):
        return model(batch[0]).pow(2).mean()

    return Trainer(model, opt, data, (n_iters, 'iteration'),
                   loss_fn=loss_fn, out=out or 'result')


def test_triggers_and_priorities():
    calls = []

    class Ext:
        def __init__(self, name, ret=None):
            self.name = name

        def __call__(self, trainer):
            calls.append((self.name, trainer.iteration))

    with tempfile.TemporaryDirectory() as d:
        t = _trainer(out=d)
        t.extend(Ext('every2'), trigger=(2, 'iteration'))
        t.extend(Ext('first'), trigger=(2, 'iteration'), priority=10)
        t.run()
    evens = [it for (n, it) in calls if n == 'every2']
    assert evens == [2, 4, 6]
    # priority=10 runs before priority=0 at each trigger point
    order = [n for (n, it) in calls if it == 2]
    assert order == ['first', 'every2']
    assert t.get_extension('every2') is not None


def test_epoch_trigger_and_stop():
    with tempfile.TemporaryDirectory() as d:
        t = _trainer(n_iters=100, out=d)
        t.stop_trigger = (2, 'epoch')
        seen = []
        t.extend(lambda tr: seen.append(tr.epoch), trigger=(1, 'epoch'),
                 name='epochly')
        t.run()
    assert t.epoch == 2
    assert seen == [1, 2]
    assert t.iteration == 6  # 3 batches x 2 epochs


def test_log_and_print_report(capsys):
    with tempfile.TemporaryDirectory() as d:
        t = _trainer(out=d)
        t.extend(LogReport(), trigger=(2, 'iteration'))
        t.extend(PrintReport(['iteration', 'main/loss']),
                 trigger=(3, 'iteration'))
        t.run()
        log = json.load(open(os.path.join(d, 'log')))
    assert [e['iteration'] for e in log] == [2, 4, 6]
    assert all('main/loss' in e and 'step_time_ms' in e for e in log)
    out = capsys.readouterr().out
    assert 'iteration=3' in out and 'iteration=6' in out


def test_state_roundtrip():
    with tempfile.TemporaryDirectory() as d:
        t = _trainer(out=d)
        t.run()
        state = t.state_dict()
        t2 = _trainer(out=d)
        t2.load_state_dict(state)
    assert t2.iteration == t.iteration
    for a, b in zip(t.model.parameters(), t2.model.parameters()):
        assert torch.equal(a, b)
