"""Distributed checkpointer with auto-resume.

Reference: chainermn/extensions/checkpoint.py (~L1-300): each rank
serializes its own trainer/optimizer/model state to
``{path}/{name}.{iteration}.{rank}``, keeps a sliding window
(garbage-collected every ``gc_interval``), and ``maybe_load`` resumes from
the newest iteration for which ALL ranks have files.  Restart with the
same world size → auto-resume (the reference's whole fault-tolerance
story; no elasticity).
"""

import os
import re

import torch


def create_multi_node_checkpointer(name, comm, cp_interval=5, gc_interval=5,
                                   path=None):
    return _MultiNodeCheckpointer(name, comm, cp_interval, gc_interval, path)


class _MultiNodeCheckpointer(object):

    name = 'MultiNodeCheckpointer'

    def __init__(self, name, comm, cp_interval, gc_interval, path):
        self._name = name
        self.comm = comm
        self.cp_interval = cp_interval
        self.gc_interval = gc_interval
        self.path = path
        self._saved = []  # iterations this rank has written
        self.save_times = []

    def _dir(self, trainer=None):
        d = self.path or (trainer.out if trainer is not None else '.')
        os.makedirs(d, exist_ok=True)
        return d

    def _file(self, directory, iteration, rank):
        return os.path.join(directory, '{}.{}.{}'.format(
            self._name, iteration, rank))

    def __call__(self, trainer):
        self.save(trainer)

    def save(self, trainer):
        import time
        t0 = time.time()
        d = self._dir(trainer)
        it = trainer.iteration
        torch.save(trainer.state_dict(), self._file(d, it, self.comm.rank))
        self._saved.append(it)
        self.save_times.append(time.time() - t0)
        if len(self._saved) >= self.gc_interval:
            self._gc(d)

    def _gc(self, directory):
        keep = set(sorted(self._saved)[-self.gc_interval:])
        for it in list(self._saved):
            if it not in keep:
                f = self._file(directory, it, self.comm.rank)
                if os.path.exists(f):
                    os.remove(f)
                self._saved.remove(it)

    def _local_iterations(self, directory):
        pat = re.compile(r'^{}\.(\d+)\.{}$'.format(
            re.escape(self._name), self.comm.rank))
        its = []
        if os.path.isdir(directory):
            for fn in os.listdir(directory):
                m = pat.match(fn)
                if m:
                    its.append(int(m.group(1)))
        return sorted(its)

    def maybe_load(self, trainer, optimizer=None):
        """Resume from the newest iteration ALL ranks have on disk."""
        d = self._dir(trainer)
        local = set(self._local_iterations(d))
        all_sets = self.comm.allgather_obj(local)
        common = set.intersection(*[set(s) for s in all_sets]) \
            if all_sets else set()
        if not common:
            return None
        it = max(common)
        state = torch.load(self._file(d, it, self.comm.rank),
                           weights_only=False)
        trainer.load_state_dict(state)
        self._saved = sorted(set(self._saved) | local)
        return it

    def get_stats(self):
        """Aggregate save timing stats at rank 0 (reference gathers via
        gather_obj)."""
        stats = self.comm.gather_obj(self.save_times, root=0)
        if self.comm.rank == 0:
            flat = [t for times in stats for t in times]
            if not flat:
                return {'count': 0}
            return {'count': len(flat),
                    'mean': sum(flat) / len(flat),
                    'max': max(flat)}
        return None

    def finalize(self):
        pass
