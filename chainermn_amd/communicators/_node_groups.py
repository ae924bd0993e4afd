"""Intra-/inter-node sub-group construction.

Reference: chainermn/communicators/_communication_utility.py
init_intra_mpi_comm / init_inter_mpi_comm / init_nccl_comm (~L60-140).
Here sub-groups are torch.distributed groups (the RCCL communicator for a
group is created lazily by torch on first use, seeded through the store —
the reference's MPI-bcast of ncclGetUniqueId).
"""

import socket

import torch.distributed as dist


def build_node_groups(comm):
    """Return (intra_group, intra_ranks, inter_group_or_None, node_lists).

    * intra_group: group of ranks on this host.
    * inter_group: group of ranks holding this rank's intra_rank across
      hosts (for intra_rank 0 this is the "local roots" group the
      hierarchical communicator reduces over).  All ranks call new_group
      for every group collectively.
    """
    host = socket.gethostname()
    entries = comm.allgather_obj((host, comm._global_rank(comm.rank)))
    host_order = []
    for h, _ in entries:
        if h not in host_order:
            host_order.append(h)
    node_lists = [[g for h, g in entries if h == hh] for hh in host_order]
    intra_group = None
    for ranks in node_lists:
        g = dist.new_group(ranks=ranks)
        if comm._global_rank(comm.rank) in ranks:
            intra_group = g
            intra_ranks = ranks
    inter_group = None
    max_intra = max(len(r) for r in node_lists)
    for i in range(max_intra):
        ranks = [r[i] for r in node_lists if len(r) > i]
        g = dist.new_group(ranks=ranks)
        if comm._global_rank(comm.rank) in ranks:
            inter_group = g
    return intra_group, intra_ranks, inter_group, node_lists
