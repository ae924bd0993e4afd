"""MultiNodeChainList: 2-rank pipelines compared against the equivalent
single-process model (reference: links_tests/test_multi_node_chain_list.py)."""

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd.links import MultiNodeChainList
from chainermn_amd.models.seq2seq import Seq2Seq, make_model_parallel
from chainermn_amd.testing import run_distributed


def _make_parts():
    torch.manual_seed(3)
    f = nn.Linear(4, 4)
    g = nn.Linear(4, 2)
    return f, g


def _pipeline_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    f, g = _make_parts()
    model = MultiNodeChainList(comm)
    if rank == 0:
        model.add_link(f, rank_in=None, rank_out=1)
    else:
        model.add_link(g, rank_in=0, rank_out=None)
    torch.manual_seed(11)
    x = torch.randn(5, 4)
    if rank == 0:
        out = model(x)
        out.backward()
        return {'gw': f.weight.grad}
    else:
        y = model()
        loss = y.pow(2).sum()
        loss.backward()
        return {'y': y.detach(), 'gw': g.weight.grad}


def test_two_stage_pipeline():
    results = run_distributed(_pipeline_worker, 2)
    f, g = _make_parts()
    torch.manual_seed(11)
    x = torch.randn(5, 4)
    y = g(f(x))
    loss = y.pow(2).sum()
    loss.backward()
    assert torch.allclose(torch.from_numpy(results[1]['y']), y.detach(),
                          atol=1e-6)
    assert torch.allclose(torch.from_numpy(results[0]['gw']), f.weight.grad,
                          atol=1e-5)
    assert torch.allclose(torch.from_numpy(results[1]['gw']), g.weight.grad,
                          atol=1e-5)


def _branching_worker(rank, size):
    """Multicast: rank0 sends to ranks 1 and 2; rank 1 and 2 feed rank 3?
    With 2 ranks here: rank0 sends the same activation twice to rank1,
    rank1 receives both and merges — exercises multi-recv + delegate
    chaining."""
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(5)
    f = nn.Linear(4, 4)
    if rank == 0:
        x = torch.randn(3, 4)
        h = f(x)
        d1 = chainermn_amd.functions.send(h, comm, 1)
        d2 = chainermn_amd.functions.send(h, comm, 1)
        d = chainermn_amd.functions.pseudo_connect(d1, d2)
        d.backward()
        return {'gw': f.weight.grad}
    else:
        a = chainermn_amd.functions.recv(comm, 0)
        b = chainermn_amd.functions.recv(comm, 0, delegate_variable=a)
        ((a + 2 * b).sum()).backward()
        return {}


def test_multicast_merge():
    results = run_distributed(_branching_worker, 2)
    torch.manual_seed(5)
    f = nn.Linear(4, 4)
    x = torch.randn(3, 4)
    h = f(x)
    ((h + 2 * h).sum()).backward()
    assert torch.allclose(torch.from_numpy(results[0]['gw']), f.weight.grad,
                          atol=1e-5)


def _seq2seq_mp_worker(rank, size):
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(1)
    model = Seq2Seq(11, 13, 16, n_layers=1)
    part = make_model_parallel(model, comm, 0, 1)
    torch.manual_seed(2)
    xs = torch.randint(0, 11, (2, 5))
    ys_in = torch.randint(0, 13, (2, 4))
    ys_out = torch.randint(0, 13, (2, 4))
    if rank == 0:
        delegate = part(xs)
        delegate.backward()
        return {'g': model.encoder.embed.weight.grad}
    else:
        loss = part(ys_in, ys_out)
        loss.backward()
        return {'loss': loss.detach(),
                'g': model.decoder.embed.weight.grad}


def test_seq2seq_model_parallel():
    results = run_distributed(_seq2seq_mp_worker, 2)
    torch.manual_seed(1)
    model = Seq2Seq(11, 13, 16, n_layers=1)
    torch.manual_seed(2)
    xs = torch.randint(0, 11, (2, 5))
    ys_in = torch.randint(0, 13, (2, 4))
    ys_out = torch.randint(0, 13, (2, 4))
    loss = model(xs, ys_in, ys_out)
    loss.backward()
    assert torch.allclose(torch.from_numpy(results[1]['loss']),
                          loss.detach(), atol=1e-5)
    assert torch.allclose(torch.from_numpy(results[0]['g']),
                          model.encoder.embed.weight.grad, atol=1e-5)
    assert torch.allclose(torch.from_numpy(results[1]['g']),
                          model.decoder.embed.weight.grad, atol=1e-5)


def _nstep_rnn_worker(rank, size):
    import torch.nn as nn
    from chainermn_amd.links import create_multi_node_n_step_rnn
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(4)
    rnn_a = nn.LSTM(6, 6, 1, batch_first=True)
    rnn_b = nn.LSTM(6, 6, 1, batch_first=True)
    torch.manual_seed(9)
    x1 = torch.randn(2, 5, 6)
    x2 = torch.randn(2, 4, 6)
    if rank == 0:
        wrapped = create_multi_node_n_step_rnn(rnn_a, comm, rank_in=None,
                                               rank_out=1)
        out, h_n, delegate = wrapped(x1)
        delegate.backward()
        return {'g': rnn_a.weight_ih_l0.grad}
    else:
        wrapped = create_multi_node_n_step_rnn(rnn_b, comm, rank_in=0,
                                               rank_out=None)
        out, h_n, _ = wrapped(x2)
        out.sum().backward()
        return {'out': out.detach(), 'g': rnn_b.weight_ih_l0.grad}


def test_multi_node_n_step_rnn():
    results = run_distributed(_nstep_rnn_worker, 2)
    import torch.nn as nn
    torch.manual_seed(4)
    rnn_a = nn.LSTM(6, 6, 1, batch_first=True)
    rnn_b = nn.LSTM(6, 6, 1, batch_first=True)
    torch.manual_seed(9)
    x1 = torch.randn(2, 5, 6)
    x2 = torch.randn(2, 4, 6)
    _, state = rnn_a(x1)
    out, _ = rnn_b(x2, state)
    out.sum().backward()
    assert torch.allclose(torch.from_numpy(results[1]['out']), out.detach(),
                          atol=1e-6)
    assert torch.allclose(torch.from_numpy(results[0]['g']),
                          rnn_a.weight_ih_l0.grad, atol=1e-5)
    assert torch.allclose(torch.from_numpy(results[1]['g']),
                          rnn_b.weight_ih_l0.grad, atol=1e-5)


def _multicast_chainlist_worker(rank, size):
    """rank 0 multicasts its output to ranks 1 and 2 (rank_out list);
    backward sums both consumers' gradients at the producer."""
    comm = chainermn_amd.create_communicator('naive')
    torch.manual_seed(8)
    f = nn.Linear(4, 4)
    g1 = nn.Linear(4, 2)
    g2 = nn.Linear(4, 2)
    model = MultiNodeChainList(comm)
    if rank == 0:
        model.add_link(f, rank_in=None, rank_out=[1, 2])
        torch.manual_seed(12)
        x = torch.randn(3, 4)
        out = model(x)
        out.backward()
        return {'gw': f.weight.grad}
    else:
        link = g1 if rank == 1 else g2
        model.add_link(link, rank_in=0, rank_out=None)
        y = model()
        (float(rank) * y.pow(2).sum()).backward()
        return {}


def test_chainlist_multicast_rank_out():
    results = run_distributed(_multicast_chainlist_worker, 3)
    torch.manual_seed(8)
    f = nn.Linear(4, 4)
    g1 = nn.Linear(4, 2)
    g2 = nn.Linear(4, 2)
    torch.manual_seed(12)
    x = torch.randn(3, 4)
    h = f(x)
    loss = 1.0 * g1(h).pow(2).sum() + 2.0 * g2(h).pow(2).sum()
    loss.backward()
    assert torch.allclose(torch.from_numpy(results[0]['gw']),
                          f.weight.grad, atol=1e-5)
