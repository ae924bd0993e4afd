"""MultiNodeChainList — graph-partition (sequential model) parallelism.

Reference: chainermn/links/multi_node_chain_list.py (~L1-230): a per-rank
view of a model graph partitioned across ranks.  ``add_link(chain,
rank_in, rank_out)`` registers a component; ``forward`` receives
activations from ``rank_in`` (ordered behind the previous component's
delegate via ``pseudo_connect``), runs the local sub-module, and sends
outputs to ``rank_out`` (int or list → multicast).  Supports non-linear
topologies (branching / merging via multiple rank_in / rank_out).

The backward graph replays sends/recvs in reverse delegate order, which is
what makes the RCCL p2p (tag-less, issue-ordered) deadlock-free.
"""

import torch.nn as nn

from chainermn_amd import functions


class MultiNodeChainList(nn.Module):

    def __init__(self, comm):
        super().__init__()
        self.comm = comm
        self._components = []  # (module_or_None, rank_in, rank_out)
        self._locals = nn.ModuleList()

    def add_link(self, link, rank_in=None, rank_out=None):
        """Register ``link`` to run on this rank when ``link`` is not None;
        other ranks register the same topology with ``link=None`` implicitly
        by only adding their own links — following the reference, every rank
        calls add_link for its OWN components only, with the global
        rank_in/rank_out wiring."""
        if link is not None:
            self._locals.append(link)
        self._components.append((link, rank_in, rank_out))
        return self

    def forward(self, *inputs):
        comm = self.comm
        delegate = None
        y = None
        for link, rank_in, rank_out in self._components:
            if link is None:
                continue
            # ---- gather inputs ----
            if rank_in is None:
                x = inputs
            else:
                rank_in_list = (rank_in if isinstance(rank_in, (list, tuple))
                                else [rank_in])
                xs = []
                for r in rank_in_list:
                    received = functions.recv(
                        comm, r, delegate_variable=delegate,
                        force_tuple=True)
                    xs.extend(received if isinstance(received, tuple)
                              else (received,))
                    # Chain subsequent recvs behind this one.
                    delegate = xs[-1]
                x = tuple(xs)
            # ---- local compute ----
            y = link(*x)
            ys = y if isinstance(y, tuple) else (y,)
            # ---- ship outputs ----
            if rank_out is not None:
                rank_out_list = (rank_out
                                 if isinstance(rank_out, (list, tuple))
                                 else [rank_out])
                for r in rank_out_list:
                    d = functions.send(ys if len(ys) > 1 else ys[0],
                                       comm, r)
                    delegate = (d if delegate is None
                                else _merge_delegates(delegate, d))
                y = delegate
        if y is None:
            raise RuntimeError(
                'rank {} has no components in this MultiNodeChainList'
                .format(comm.rank))
        return y


def _merge_delegates(a, b):
    """Combine two delegate variables into one graph edge (backward reaches
    both, b's subgraph first — reverse of forward issue order)."""
    from chainermn_amd.functions import pseudo_connect
    return pseudo_connect(a, b)
