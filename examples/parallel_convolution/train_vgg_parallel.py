#!/usr/bin/env python3
"""Channel-parallel VGG — the tensor-parallel-style demo
(reference: examples/parallel_convolution/): each rank owns a slice of
every conv layer's OUTPUT channels; after each conv the activations are
all-gathered (differentiable) and concatenated, so every rank sees the
full feature map for the next layer.

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 examples/parallel_convolution/train_vgg_parallel.py
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))  # repo root (in-tree package)


import argparse
import time

import torch
import torch.nn as nn

import chainermn_amd
from chainermn_amd import functions


class ParallelConv2d(nn.Module):
    """Conv2d with out_channels split across the communicator."""

    def __init__(self, comm, in_ch, out_ch, **kw):
        super().__init__()
        assert out_ch % comm.size == 0, 'out_ch must divide comm size'
        self.comm = comm
        self.conv = nn.Conv2d(in_ch, out_ch // comm.size, **kw)

    def forward(self, x):
        local = self.conv(x)
        if self.comm.size == 1:
            return local
        parts = functions.allgather(self.comm, local.contiguous())
        return torch.cat(parts, dim=1)


class ParallelVGG(nn.Module):
    """VGG-ish stack with channel-parallel convolutions."""

    def __init__(self, comm, num_classes=10):
        super().__init__()
        cfg = [(3, 64), (64, 128), (128, 256), (256, 256)]
        layers = []
        for cin, cout in cfg:
            layers += [ParallelConv2d(comm, cin, cout, kernel_size=3,
                                      padding=1),
                       nn.ReLU(inplace=True), nn.MaxPool2d(2)]
        self.features = nn.Sequential(*layers)
        self.head = nn.Linear(256 * 2 * 2, num_classes)

    def forward(self, x):
        h = self.features(x)
        return self.head(h.flatten(1))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batchsize', type=int, default=64)
    ap.add_argument('--iterations', type=int, default=50)
    args = ap.parse_args()

    chainermn_amd.global_except_hook.add_hook()
    comm = chainermn_amd.create_communicator(
        None if torch.cuda.is_available() else 'naive')
    device = torch.device('cuda', comm.intra_rank) \
        if torch.cuda.is_available() else torch.device('cpu')
    torch.manual_seed(11)  # identical conv slices start identical per rank
    model = ParallelVGG(comm).to(device)
    # channel-parallel: each rank's conv params are DISTINCT shards; only
    # the shared head is data-parallel — so use a plain optimizer and
    # all-reduce only the head grads.
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    loss_fn = nn.CrossEntropyLoss()
    t0 = time.time()
    for it in range(args.iterations):
        g = torch.Generator().manual_seed(100 + it)  # same batch everywhere
        x = torch.rand(args.batchsize, 3, 32, 32, generator=g).to(device)
        y = (torch.arange(args.batchsize) % 10).to(device)
        opt.zero_grad()
        loss = loss_fn(model(x), y)
        loss.backward()
        # head is replicated: average its grads to stay in sync
        if comm.size > 1:
            import torch.distributed as dist
            for p in model.head.parameters():
                dist.all_reduce(p.grad, group=comm.group)
                p.grad /= comm.size
        opt.step()
        if comm.rank == 0 and (it + 1) % 10 == 0:
            print('iter {}: loss={:.3f} ({:.1f} it/s)'.format(
                it + 1, float(loss), (it + 1) / (time.time() - t0)),
                flush=True)


if __name__ == '__main__':
    main()
