#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 synthetic-ImageNet data-parallel training.

Metric (BASELINE.json): images/sec whole-node, ResNet-50, synthetic
3x224x224, random-init weights; scaling efficiency at 1/2/4/8 MI355X GPUs.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Training step = forward + backward + gradient all-reduce + optimizer
update (nothing skipped): bf16 model in channels_last, RCCL bf16
all-reduce of the flat grad arena overlapped on a side stream
(double-buffered, 1-step stale — the reference's flagship config), fused
CDNA4 SGD-momentum update with fp32 master weights.
"""

import argparse
import json
import os
import sys
import time

# MIOpen: fresh boxes have no tuning DB; FAST find avoids multi-minute
# kernel search during warmup.  Must be set before the first conv.
os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')

import torch  # noqa: E402


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=20)
    ap.add_argument('--warmup', type=int, default=8)
    ap.add_argument('--batch-size', type=int, default=256,
                    help='per-GPU batch (weak scaling)')
    ap.add_argument('--model', default='resnet50',
                    choices=['resnet50', 'mlp'])
    ap.add_argument('--dtype', default='bf16', choices=['bf16', 'fp32'])
    ap.add_argument('--optimizer', default='sgd', choices=['sgd', 'lars'])
    ap.add_argument('--no-overlap', action='store_true',
                    help='synchronous all-reduce (no double buffering)')
    ap.add_argument('--mnbn', action='store_true',
                    help='MultiNodeBatchNormalization instead of local BN')
    ap.add_argument('--communicator', default='pure_rccl')
    return ap.parse_args()


def main():
    args = parse_args()
    use_gpu = torch.cuda.is_available()
    world = int(os.environ.get('WORLD_SIZE', '1'))
    import chainermn_amd
    from chainermn_amd.models import MLP, ResNet50
    from chainermn_amd.parallel import (FusedDataParallelLARS,
                                        FusedDataParallelSGD)

    if world > 1 or 'RANK' in os.environ:
        comm = chainermn_amd.create_communicator(
            args.communicator if use_gpu else 'naive')
    else:
        comm = chainermn_amd.create_communicator('dummy')
    rank = comm.rank
    n_gpus = comm.size

    device = torch.device('cuda', torch.cuda.current_device()) if use_gpu \
        else torch.device('cpu')
    dtype = torch.bfloat16 if args.dtype == 'bf16' else torch.float32
    torch.manual_seed(1234)  # identical random init on every rank

    if args.model == 'resnet50':
        if args.mnbn:
            from chainermn_amd.links import MultiNodeBatchNormalization
            model = ResNet50(norm_layer=lambda c: MultiNodeBatchNormalization(
                c, comm))
        else:
            model = ResNet50()
        model = model.to(device=device, dtype=dtype,
                         memory_format=torch.channels_last)
        x = torch.randn(args.batch_size, 3, 224, 224, device=device,
                        dtype=dtype).to(memory_format=torch.channels_last)
    else:
        model = MLP().to(device=device, dtype=dtype)
        x = torch.randn(args.batch_size, 784, device=device, dtype=dtype)
    y = torch.randint(0, 1000 if args.model == 'resnet50' else 10,
                      (args.batch_size,), device=device)
    model.train()

    overlap = (not args.no_overlap) and use_gpu
    if use_gpu:
        cls = FusedDataParallelLARS if args.optimizer == 'lars' \
            else FusedDataParallelSGD
        opt = cls(model, comm, lr=0.05 * n_gpus, momentum=0.9,
                  weight_decay=1e-4 if args.optimizer == 'sgd' else 5e-5,
                  overlap=overlap)
    else:
        opt = chainermn_amd.create_multi_node_optimizer(
            torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9),
            comm).setup(model, broadcast=n_gpus > 1)
    loss_fn = torch.nn.CrossEntropyLoss()

    def step():
        opt.zero_grad(set_to_none=False)
        out = model(x)
        loss = loss_fn(out.float(), y)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines throughput)
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.size > 1:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    global_batch = args.batch_size * n_gpus
    images = global_batch * args.steps
    value = images / elapsed
    if rank == 0:
        result = {
            'metric': 'images/sec (whole node) ResNet-50 synthetic ImageNet'
                      if args.model == 'resnet50' else 'images/sec MLP',
            'value': round(value, 2),
            'unit': 'images/sec',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(elapsed / args.steps * 1000, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': args.dtype,
            'data': 'synthetic',
            'config': {
                'model': args.model,
                'global_batch': global_batch,
                'image': '3x224x224' if args.model == 'resnet50' else '784',
                'parallelism': 'dp{}'.format(n_gpus),
                'optimizer': args.optimizer,
                'overlap': overlap,
                'allreduce_dtype': args.dtype,
                'mnbn': args.mnbn,
            },
        }
        print(json.dumps(result), flush=True)
    return 0


if __name__ == '__main__':
    sys.exit(main())
