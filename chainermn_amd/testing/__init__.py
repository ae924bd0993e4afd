"""Test helpers: run a function in an N-process torch.distributed world.

Replaces the reference's ``mpiexec -np 2 pytest`` harness:
multi-node is emulated by N processes on one host.  CPU worlds use gloo
with a FileStore (fork start method — fast, no re-import); GPU worlds use
spawn + RCCL with a TCP store on 127.0.0.1.
"""

import multiprocessing
import os
import sys
import tempfile
import traceback

import torch
import torch.distributed as dist


def _detensor(obj):
    """Tensors -> numpy for queue transport (torch's ForkingPickler ships
    tensors via shared memory whose lifetime ends with the worker)."""
    if isinstance(obj, torch.Tensor):
        return obj.detach().cpu().numpy()
    if isinstance(obj, dict):
        return {k: _detensor(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_detensor(v) for v in obj)
    return obj


def _worker(rank, nprocs, store_file, fn, args, backend, errq, outq):
    try:
        # forked workers must not re-enter the parent's OpenMP pool
        torch.set_num_threads(1)
        os.environ['RANK'] = str(rank)
        os.environ['WORLD_SIZE'] = str(nprocs)
        os.environ['MASTER_ADDR'] = '127.0.0.1'
        if backend != 'gloo':
            torch.cuda.set_device(rank)
        store = dist.FileStore(store_file, nprocs)
        dist.init_process_group(backend, store=store, rank=rank,
                                world_size=nprocs)
        result = fn(rank, nprocs, *args)
        outq.put((rank, _detensor(result)))
        dist.destroy_process_group()
    except Exception:
        errq.put((rank, traceback.format_exc()))
        sys.exit(1)


def run_distributed(fn, nprocs=2, args=(), backend='gloo', timeout=180,
                    method=None):
    """Run ``fn(rank, nprocs, *args)`` in ``nprocs`` processes.

    Returns [result_rank0, result_rank1, ...].  Raises with the failing
    rank's traceback on any error.
    """
    if method is None:
        method = 'fork' if backend == 'gloo' else 'spawn'
    ctx = multiprocessing.get_context(method)
    errq = ctx.Queue()
    outq = ctx.Queue()
    fd, store_file = tempfile.mkstemp(prefix='cmnamd_store_')
    os.close(fd)
    os.unlink(store_file)  # FileStore creates it
    procs = []
    for rank in range(nprocs):
        p = ctx.Process(target=_worker,
                        args=(rank, nprocs, store_file, fn, args, backend,
                              errq, outq))
        p.start()
        procs.append(p)
    results = {}
    errors = []
    import queue as queue_mod
    import time
    deadline = time.time() + timeout
    pending = set(range(nprocs))
    while pending and time.time() < deadline:
        try:
            rank, res = outq.get(timeout=0.2)
            results[rank] = res
            pending.discard(rank)
        except queue_mod.Empty:
            pass
        while not errq.empty():
            errors.append(errq.get())
            pending.discard(errors[-1][0])
        if errors:
            break
    for p in procs:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
            p.join(timeout=5)
    try:
        os.unlink(store_file)
    except OSError:
        pass
    while not errq.empty():
        errors.append(errq.get())
    if errors:
        msgs = '\n'.join('--- rank {} ---\n{}'.format(r, tb)
                         for r, tb in errors)
        raise RuntimeError('distributed test failed:\n' + msgs)
    if pending:
        raise RuntimeError(
            'distributed test timed out; missing ranks {}'.format(pending))
    return [results[r] for r in range(nprocs)]


def get_device(use_gpu=False):
    if use_gpu and torch.cuda.is_available():
        return torch.device('cuda', torch.cuda.current_device())
    return torch.device('cpu')
