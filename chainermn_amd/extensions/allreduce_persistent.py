"""AllreducePersistent — reconcile persistent values (BN running stats)
across ranks.

Reference: chainermn/extensions/allreduce_persistent.py (~L1-50): all-
reduce-average every persistent value so plain (non-MNBN) BatchNorm
statistics agree across workers before evaluation/snapshot.  Torch
equivalent of "persistent values" = module buffers.
"""

import torch
import torch.distributed as dist


class AllreducePersistent(object):

    name = 'AllreducePersistent'

    def __init__(self, model, comm):
        self.model = model
        self.comm = comm

    def __call__(self, trainer=None):
        size = self.comm.size
        if size == 1:
            return
        with torch.no_grad():
            for b in self.model.buffers():
                if not b.dtype.is_floating_point:
                    continue
                dist.all_reduce(b.data, group=self.comm.group)
                b.data.div_(size)
