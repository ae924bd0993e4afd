#!/usr/bin/env python3
"""Seq2seq LSTM training, data-parallel or model-parallel
(reference: examples/seq2seq/seq2seq.py and its _mp variants).

Model parallel (encoder on rank 0, decoder on rank 1; hidden states cross
xGMI via differentiable RCCL send/recv):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 examples/seq2seq/seq2seq.py --model-parallel

Synthetic copy-task data (no network access).
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))  # repo root (in-tree package)


import argparse
import time

import torch

import chainermn_amd
from chainermn_amd.models.seq2seq import Seq2Seq, make_model_parallel

VOCAB = 64


def synthetic_batch(batch, src_len, tgt_len, device, seed):
    g = torch.Generator().manual_seed(seed)
    xs = torch.randint(2, VOCAB, (batch, src_len), generator=g).to(device)
    ys = torch.randint(2, VOCAB, (batch, tgt_len), generator=g).to(device)
    ys_in = torch.cat([torch.zeros(batch, 1, dtype=torch.long,
                                   device=device), ys[:, :-1]], dim=1)
    return xs, ys_in, ys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--communicator', default=None)
    ap.add_argument('--batchsize', type=int, default=64)
    ap.add_argument('--unit', type=int, default=512)
    ap.add_argument('--layer', type=int, default=2)
    ap.add_argument('--iterations', type=int, default=100)
    ap.add_argument('--model-parallel', action='store_true')
    args = ap.parse_args()

    chainermn_amd.global_except_hook.add_hook()
    comm = chainermn_amd.create_communicator(args.communicator)
    device = torch.device('cuda', comm.intra_rank) \
        if torch.cuda.is_available() else torch.device('cpu')
    torch.manual_seed(3)
    model = Seq2Seq(VOCAB, VOCAB, args.unit, args.layer).to(device)

    if args.model_parallel:
        assert comm.size == 2, 'model-parallel demo uses exactly 2 ranks'
        part = make_model_parallel(model, comm, 0, 1)
        opt = torch.optim.Adam(part.parameters(), lr=1e-3)
        t0 = time.time()
        for it in range(args.iterations):
            xs, ys_in, ys = synthetic_batch(args.batchsize, 20, 15, device,
                                            seed=it)
            opt.zero_grad()
            if comm.rank == 0:
                delegate = part(xs)
                delegate.backward()
            else:
                loss = part(ys_in, ys)
                loss.backward()
            opt.step()
            if comm.rank == 1 and (it + 1) % 20 == 0:
                print('iter {}: loss={:.3f} ({:.1f} it/s)'.format(
                    it + 1, float(loss), (it + 1) / (time.time() - t0)),
                    flush=True)
    else:
        opt = chainermn_amd.create_multi_node_optimizer(
            torch.optim.Adam(model.parameters(), lr=1e-3), comm)
        opt.setup(model, broadcast=comm.size > 1)
        for it in range(args.iterations):
            xs, ys_in, ys = synthetic_batch(args.batchsize, 20, 15, device,
                                            seed=it * comm.size + comm.rank)
            opt.zero_grad(set_to_none=False)
            loss = model(xs, ys_in, ys)
            loss.backward()
            opt.step()
            if comm.rank == 0 and (it + 1) % 20 == 0:
                print('iter {}: loss={:.3f}'.format(it + 1, float(loss)),
                      flush=True)


if __name__ == '__main__':
    main()
