"""Process bootstrap without MPI.

The reference used ``mpiexec`` + ``mpi4py.MPI.COMM_WORLD`` as the only
launcher and derived intra/inter-node geometry via
``MPI_Comm_split_type(SHARED)`` (reference:
chainermn/communicators/_communication_utility.py init_ranks, ~L20-60).

On MI355X there is no MPI: processes are launched one-per-GPU with
``torchrun`` (or any launcher exporting RANK/WORLD_SIZE/MASTER_ADDR/
MASTER_PORT), rendezvous goes through ``torch.distributed``'s TCPStore, and
node membership is computed by all-gathering hostnames over the gloo side
channel.
"""

import datetime
import os
import socket

import torch
import torch.distributed as dist


def free_port() -> int:
    """Bind an ephemeral port on 127.0.0.1 and return it."""
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def default_backend() -> str:
    # 'cuda:nccl' is RCCL on ROCm; the cpu:gloo component carries the
    # pickled-object control plane (the reference's MPI *_obj channel).
    if torch.cuda.is_available():
        return 'cpu:gloo,cuda:nccl'
    return 'gloo'


def ensure_initialized(backend=None, timeout_s=600):
    """Initialise the default process group if not already done.

    Reads RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT from the environment
    (torchrun-compatible).  Falls back to a single-process world so that
    library code (and DummyCommunicator-free single-rank runs) work without
    a launcher.
    """
    if dist.is_available() and dist.is_initialized():
        return
    if 'RANK' not in os.environ or 'WORLD_SIZE' not in os.environ:
        os.environ.setdefault('RANK', '0')
        os.environ.setdefault('WORLD_SIZE', '1')
        os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
        os.environ.setdefault('MASTER_PORT', str(free_port()))
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29511')
    if backend is None:
        backend = default_backend()
    kwargs = {}
    if torch.cuda.is_available():
        local_rank = int(os.environ.get(
            'LOCAL_RANK', int(os.environ['RANK']) % max(torch.cuda.device_count(), 1)))
        torch.cuda.set_device(local_rank)
        kwargs['device_id'] = torch.device('cuda', local_rank)
    dist.init_process_group(
        backend=backend,
        timeout=datetime.timedelta(seconds=timeout_s),
        **kwargs)


def get_topology(group=None):
    """Return (intra_rank, intra_size, inter_rank, inter_size, local_ranks).

    Node membership by hostname, mirroring the reference's
    ``MPI_Comm_split_type(COMM_TYPE_SHARED)`` geometry:
    ``intra_rank`` = index among same-host ranks ordered by global rank
    (by convention the GPU device id), ``inter_rank`` = index of this host.
    """
    rank = dist.get_rank(group)
    size = dist.get_world_size(group)
    host = socket.gethostname()
    hosts = [None] * size
    dist.all_gather_object(hosts, host, group=group)
    # Hosts ordered by first appearance.
    host_order = []
    for h in hosts:
        if h not in host_order:
            host_order.append(h)
    local_ranks = [r for r, h in enumerate(hosts) if h == host]
    intra_rank = local_ranks.index(rank)
    intra_size = len(local_ranks)
    inter_rank = host_order.index(host)
    inter_size = len(host_order)
    return intra_rank, intra_size, inter_rank, inter_size, local_ranks
