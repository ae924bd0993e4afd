"""Multi-node N-step RNN — hidden states flow across ranks.

Reference: chainermn/links/n_step_rnn.py (~L1-90):
``create_multi_node_n_step_rnn(actual_rnn, comm, rank_in, rank_out)``
wraps a recurrent module so its initial hidden state is received from
``rank_in`` and its final hidden state is sent to ``rank_out`` — the
encoder/decoder split of seq2seq across GPUs (BASELINE.json config 4).

Works with ``torch.nn.LSTM``/``GRU``/``RNN`` (batch_first or not): the
wrapped call is ``(output, h_n)`` / ``(output, (h_n, c_n))``.
"""

import torch
import torch.nn as nn

from chainermn_amd import functions


class MultiNodeNStepRNN(nn.Module):

    def __init__(self, actual_rnn, comm, rank_in=None, rank_out=None):
        super().__init__()
        self.actual_rnn = actual_rnn
        self.comm = comm
        self.rank_in = rank_in
        self.rank_out = rank_out

    def forward(self, x, hx=None):
        comm = self.comm
        delegate = None
        if self.rank_in is not None:
            received = functions.recv(comm, self.rank_in, force_tuple=True)
            hx = received[0] if len(received) == 1 else tuple(received)
            delegate = received[0]
        out, h_n = self.actual_rnn(x, hx)
        if self.rank_out is not None:
            payload = h_n if isinstance(h_n, torch.Tensor) else tuple(h_n)
            d = functions.send(payload, comm, self.rank_out)
            if delegate is not None:
                d = functions.pseudo_connect(delegate, d)
            return out, h_n, d
        return out, h_n, None


def create_multi_node_n_step_rnn(actual_rnn, comm, rank_in=None,
                                 rank_out=None):
    return MultiNodeNStepRNN(actual_rnn, comm, rank_in=rank_in,
                             rank_out=rank_out)
