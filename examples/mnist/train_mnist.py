#!/usr/bin/env python3
"""Data-parallel MNIST MLP — the canonical plumbing example
(reference: examples/mnist/train_mnist.py).

Launch:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 examples/mnist/train_mnist.py --epoch 2

No network access: uses a synthetic MNIST-shaped dataset (random images,
balanced labels).  Swap in a real dataset loader where available.
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))  # repo root (in-tree package)


import argparse

import torch

import chainermn_amd
from chainermn_amd.datasets import scatter_dataset
from chainermn_amd.extensions import (ObservationAggregator,
                                      create_multi_node_checkpointer)
from chainermn_amd.models import MLP
from chainermn_amd.training import Evaluator, LogReport, PrintReport, Trainer


def synthetic_mnist(n=4096, seed=0):
    g = torch.Generator().manual_seed(seed)
    xs = torch.rand(n, 784, generator=g)
    ys = torch.arange(n) % 10
    # clone(): pickling a VIEW serializes the whole backing storage
    # (12 MB per item here) — clone keeps shards compact
    return [(xs[i].clone(), int(ys[i])) for i in range(n)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--communicator', default=None)
    ap.add_argument('--batchsize', type=int, default=100)
    ap.add_argument('--epoch', type=int, default=2)
    ap.add_argument('--lr', type=float, default=0.05)
    ap.add_argument('--checkpoint', action='store_true')
    args = ap.parse_args()

    chainermn_amd.global_except_hook.add_hook()
    comm = chainermn_amd.create_communicator(args.communicator)
    device = torch.device('cuda', comm.intra_rank) \
        if torch.cuda.is_available() else torch.device('cpu')

    model = MLP().to(device)
    optimizer = chainermn_amd.create_multi_node_optimizer(
        torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9), comm)
    optimizer.setup(model)

    train = scatter_dataset(synthetic_mnist(4096), comm, shuffle=True,
                            seed=42)
    val = scatter_dataset(synthetic_mnist(512, seed=1), comm)

    def batches(data, bs):
        for i in range(0, len(data), bs):
            chunk = data[i:i + bs]
            yield (torch.stack([x for x, _ in chunk]),
                   torch.tensor([y for _, y in chunk]))

    def loss_fn(model, batch):
        x, y = batch
        return torch.nn.functional.cross_entropy(model(x), y)

    class EpochIter:
        def __iter__(self):
            return batches(train, args.batchsize)

        def __len__(self):
            return (len(train) + args.batchsize - 1) // args.batchsize

    trainer = Trainer(model, optimizer, EpochIter(), (args.epoch, 'epoch'),
                      loss_fn=loss_fn, device=device, out='result_mnist')
    evaluator = chainermn_amd.create_multi_node_evaluator(
        Evaluator(batches(val, args.batchsize), model, loss_fn=loss_fn,
                  device=device), comm)
    trainer.extend(evaluator, trigger=(1, 'epoch'))
    trainer.extend(ObservationAggregator(comm, 'main/loss'),
                   trigger=(10, 'iteration'))
    if args.checkpoint:
        cp = create_multi_node_checkpointer('mnist', comm, path='result_mnist')
        cp.maybe_load(trainer)
        trainer.extend(cp, trigger=(100, 'iteration'))
    if comm.rank == 0:
        trainer.extend(LogReport(), trigger=(10, 'iteration'))
        trainer.extend(PrintReport(['iteration', 'main/loss']),
                       trigger=(10, 'iteration'))
    trainer.run()
    if comm.rank == 0:
        print('done: iteration={} loss={:.4f}'.format(
            trainer.iteration, trainer.observation.get('main/loss')))


if __name__ == '__main__':
    main()
