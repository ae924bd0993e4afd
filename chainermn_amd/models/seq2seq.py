"""Seq2seq LSTM encoder-decoder (BASELINE.json config 4).

Reference analog: examples/seq2seq/seq2seq.py — an LSTM encoder-decoder
whose model-parallel variant splits encoder and decoder across ranks via
``create_multi_node_n_step_rnn`` / ``MultiNodeChainList``.

``Seq2Seq`` here is the single-process model; ``make_model_parallel``
returns (encoder_part, decoder_part) wrappers for a 2-rank split where the
final hidden state crosses the xGMI link via differentiable send/recv.
"""

import torch.nn as nn

from chainermn_amd import functions


class Encoder(nn.Module):

    def __init__(self, n_vocab, n_units, n_layers=2):
        super().__init__()
        self.embed = nn.Embedding(n_vocab, n_units)
        self.rnn = nn.LSTM(n_units, n_units, n_layers, batch_first=True)

    def forward(self, xs):
        _, state = self.rnn(self.embed(xs))
        return state  # (h_n, c_n)


class Decoder(nn.Module):

    def __init__(self, n_vocab, n_units, n_layers=2):
        super().__init__()
        self.embed = nn.Embedding(n_vocab, n_units)
        self.rnn = nn.LSTM(n_units, n_units, n_layers, batch_first=True)
        self.out = nn.Linear(n_units, n_vocab)

    def forward(self, ys_in, state):
        h, _ = self.rnn(self.embed(ys_in), state)
        return self.out(h)


class Seq2Seq(nn.Module):
    """Teacher-forced training model; forward returns mean CE loss."""

    def __init__(self, n_src_vocab, n_tgt_vocab, n_units, n_layers=2):
        super().__init__()
        self.encoder = Encoder(n_src_vocab, n_units, n_layers)
        self.decoder = Decoder(n_tgt_vocab, n_units, n_layers)
        self.loss_fn = nn.CrossEntropyLoss()

    def forward(self, xs, ys_in, ys_out):
        state = self.encoder(xs)
        logits = self.decoder(ys_in, state)
        return self.loss_fn(logits.reshape(-1, logits.shape[-1]),
                            ys_out.reshape(-1))


class EncoderPart(nn.Module):
    """Rank holding the encoder: encode, send (h, c) to decoder rank."""

    def __init__(self, encoder, comm, decoder_rank):
        super().__init__()
        self.encoder = encoder
        self.comm = comm
        self.decoder_rank = decoder_rank

    def forward(self, xs):
        h, c = self.encoder(xs)
        delegate = functions.send((h.contiguous(), c.contiguous()),
                                  self.comm, self.decoder_rank)
        return delegate  # call .backward() on this


class DecoderPart(nn.Module):
    """Rank holding the decoder: recv (h, c), decode, return loss."""

    def __init__(self, decoder, comm, encoder_rank):
        super().__init__()
        self.decoder = decoder
        self.comm = comm
        self.encoder_rank = encoder_rank
        self.loss_fn = nn.CrossEntropyLoss()

    def forward(self, ys_in, ys_out):
        h, c = functions.recv(self.comm, self.encoder_rank,
                              force_tuple=True)
        logits = self.decoder(ys_in, (h, c))
        return self.loss_fn(logits.reshape(-1, logits.shape[-1]),
                            ys_out.reshape(-1))


def make_model_parallel(model, comm, encoder_rank=0, decoder_rank=1):
    """Split a Seq2Seq across two ranks; returns this rank's part."""
    if comm.rank == encoder_rank:
        return EncoderPart(model.encoder, comm, decoder_rank)
    elif comm.rank == decoder_rank:
        return DecoderPart(model.decoder, comm, encoder_rank)
    raise ValueError('rank {} not part of this 2-way split'.format(comm.rank))
