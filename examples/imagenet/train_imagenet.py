#!/usr/bin/env python3
"""Data-parallel ResNet-50 training — the benchmark workload
(reference: examples/imagenet/train_imagenet.py).

Launch (8 GPUs):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/imagenet/train_imagenet.py

Uses synthetic ImageNet-shaped data (no network access).  The hot path is
the MI355X-native one: bf16 params/grads in flat arenas, overlapped RCCL
all-reduce, fused CDNA4 optimizer (see chainermn_amd/parallel/fused_dp.py)
— `bench.py` at the repo root is the measured version of this script.
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))  # repo root (in-tree package)


import argparse
import time

import torch

import chainermn_amd
from chainermn_amd.models import ResNet50


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--communicator', default=None)
    ap.add_argument('--batchsize', type=int, default=256)
    ap.add_argument('--iterations', type=int, default=100)
    ap.add_argument('--lr', type=float, default=0.05)
    ap.add_argument('--optimizer', default='sgd', choices=['sgd', 'lars'])
    ap.add_argument('--mnbn', action='store_true')
    args = ap.parse_args()

    chainermn_amd.global_except_hook.add_hook()
    comm = chainermn_amd.create_communicator(args.communicator)
    use_gpu = torch.cuda.is_available()
    device = torch.device('cuda', comm.intra_rank) if use_gpu \
        else torch.device('cpu')
    torch.manual_seed(7)

    if args.mnbn and use_gpu:
        model = ResNet50(fused_ops=True, comm=comm)  # fused multi-node BN
    elif args.mnbn:
        from chainermn_amd.links import MultiNodeBatchNormalization
        model = ResNet50(norm_layer=lambda c: MultiNodeBatchNormalization(
            c, comm))
    elif use_gpu:
        model = ResNet50(fused_ops=True)
    else:
        model = ResNet50()
    dtype = torch.bfloat16 if use_gpu else torch.float32
    model = model.to(device=device, dtype=dtype,
                     memory_format=torch.channels_last)

    if use_gpu:
        from chainermn_amd.parallel import (FusedDataParallelLARS,
                                            FusedDataParallelSGD)
        cls = FusedDataParallelLARS if args.optimizer == 'lars' \
            else FusedDataParallelSGD
        opt = cls(model, comm, lr=args.lr * comm.size, momentum=0.9)
    else:
        opt = chainermn_amd.create_multi_node_optimizer(
            torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9),
            comm).setup(model, broadcast=comm.size > 1)

    x = torch.randn(args.batchsize, 3, 224, 224, device=device, dtype=dtype
                    ).to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batchsize,), device=device)
    loss_fn = torch.nn.CrossEntropyLoss()
    model.train()

    t0 = time.time()
    for it in range(args.iterations):
        opt.zero_grad(set_to_none=False)
        loss = loss_fn(model(x).float(), y)
        loss.backward()
        opt.step()
        if comm.rank == 0 and (it + 1) % 20 == 0:
            if use_gpu:
                torch.cuda.synchronize()
            dt = time.time() - t0
            print('iter {}: loss={:.3f} {:.1f} img/s (whole job)'.format(
                it + 1, float(loss), args.batchsize * comm.size * (it + 1)
                / dt), flush=True)


if __name__ == '__main__':
    main()
