"""Minimal trainer loop — the host-framework surface the reference's
extensions plugged into.

The reference relied on Chainer's ``Trainer``/``Extension``/``Reporter``
protocol (outside its own repo).  This module provides the minimal
equivalent so the multi-node extensions (checkpointer, observation
aggregator, multi-node snapshot, evaluator) have the same integration
points: extensions are callables invoked on a trigger with access to the
trainer state; observations accumulate per iteration in
``trainer.observation``.
"""

import os
import time

import torch


def _match_trigger(trainer, trigger):
    if trigger is None:
        return False
    n, unit = trigger
    if unit == 'iteration':
        return trainer.iteration % n == 0
    if unit == 'epoch':
        return trainer.is_new_epoch and trainer.epoch % n == 0
    raise ValueError('trigger unit must be iteration or epoch')


class _ExtensionEntry(object):

    def __init__(self, ext, trigger, name, priority):
        self.ext = ext
        self.trigger = trigger
        self.name = name
        self.priority = priority


class Trainer(object):
    """Drives ``optimizer.update`` over an iterator until stop_trigger.

    ``model(batch...) -> loss`` or a custom ``loss_fn(model, batch)``.
    """

    def __init__(self, model, optimizer, train_iter, stop_trigger,
                 loss_fn=None, device=None, out='result'):
        self.model = model
        self.optimizer = optimizer
        self.train_iter = train_iter
        self.stop_trigger = stop_trigger  # (n, 'iteration'|'epoch')
        self.loss_fn = loss_fn
        self.device = device
        self.out = out
        self.iteration = 0
        self.epoch = 0
        self.is_new_epoch = False
        self.observation = {}
        self.observations_history = []
        self._extensions = []
        self._start_time = None

    # -- extension protocol ------------------------------------------------
    def extend(self, extension, trigger=(1, 'iteration'), name=None,
               priority=0):
        name = name or getattr(extension, 'name', None) \
            or extension.__class__.__name__
        self._extensions.append(
            _ExtensionEntry(extension, trigger, name, priority))
        self._extensions.sort(key=lambda e: -e.priority)
        return self

    def get_extension(self, name):
        for e in self._extensions:
            if e.name == name:
                return e.ext
        raise KeyError(name)

    # -- state (for checkpoint/resume) ------------------------------------
    def state_dict(self):
        state = {
            'iteration': self.iteration,
            'epoch': self.epoch,
            'model': self.model.state_dict(),
        }
        if hasattr(self.optimizer, 'state_dict'):
            state['optimizer'] = self.optimizer.state_dict()
        return state

    def load_state_dict(self, state):
        self.iteration = state['iteration']
        self.epoch = state['epoch']
        self.model.load_state_dict(state['model'])
        if 'optimizer' in state and hasattr(self.optimizer,
                                            'load_state_dict'):
            self.optimizer.load_state_dict(state['optimizer'])

    # -- main loop ---------------------------------------------------------
    def _stop(self):
        n, unit = self.stop_trigger
        if unit == 'iteration':
            return self.iteration >= n
        return self.epoch >= n

    def step(self, batch):
        import time as _time
        _t0 = _time.perf_counter()
        self.observation = {}
        model = self.model
        if self.loss_fn is not None:
            loss = self.loss_fn(model, batch)
        else:
            if isinstance(batch, (tuple, list)):
                loss = model(*batch)
            else:
                loss = model(batch)
        self.optimizer.zero_grad(set_to_none=False)
        loss.backward()
        self.optimizer.step()
        self.iteration += 1
        self.observation.setdefault('main/loss', float(loss.detach()))
        self.observation['iteration'] = self.iteration
        self.observation['epoch'] = self.epoch
        self.observation['step_time_ms'] = (
            (_time.perf_counter() - _t0) * 1000)
        for e in self._extensions:
            if _match_trigger(self, e.trigger):
                e.ext(self)
        self.observations_history.append(dict(self.observation))
        return loss

    def run(self):
        if self.out:
            os.makedirs(self.out, exist_ok=True)
        self._start_time = time.time()
        while not self._stop():
            self.is_new_epoch = False
            for batch in self.train_iter:
                if self.device is not None:
                    batch = _to_device(batch, self.device)
                self.step(batch)
                if self._stop():
                    return
            self.epoch += 1
            self.is_new_epoch = True
            for e in self._extensions:
                if e.trigger and e.trigger[1] == 'epoch' \
                        and _match_trigger(self, e.trigger):
                    e.ext(self)

    @property
    def elapsed_time(self):
        return time.time() - (self._start_time or time.time())


def _to_device(batch, device):
    if isinstance(batch, torch.Tensor):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, (tuple, list)):
        return type(batch)(_to_device(b, device) for b in batch)
    if isinstance(batch, dict):
        return {k: _to_device(v, device) for k, v in batch.items()}
    return batch


class Evaluator(object):
    """Local evaluator: mean loss/metrics over a validation iterable."""

    name = 'validation'

    def __init__(self, iterator, model, loss_fn=None, device=None,
                 metrics=None):
        self.iterator = iterator
        self.model = model
        self.loss_fn = loss_fn
        self.device = device
        self.metrics = metrics or {}

    def evaluate(self):
        model = self.model
        was_training = getattr(model, 'training', False)
        model.eval()
        totals = {}
        count = 0
        with torch.no_grad():
            for batch in self.iterator:
                if self.device is not None:
                    batch = _to_device(batch, self.device)
                if self.loss_fn is not None:
                    loss = self.loss_fn(model, batch)
                else:
                    loss = model(*batch) if isinstance(batch, (tuple, list)) \
                        else model(batch)
                totals['validation/main/loss'] = totals.get(
                    'validation/main/loss', 0.0) + float(loss)
                for k, fn in self.metrics.items():
                    totals['validation/main/' + k] = totals.get(
                        'validation/main/' + k, 0.0) + float(fn(model, batch))
                count += 1
        if was_training:
            model.train()
        return {k: v / max(count, 1) for k, v in totals.items()}

    def __call__(self, trainer):
        result = self.evaluate()
        trainer.observation.update(result)
        return result


class LogReport(object):
    """Append observations to trainer.out/log (rank-0 idiom: wrap in
    ``if comm.rank == 0``, as every reference example did)."""

    name = 'LogReport'

    def __init__(self, filename='log'):
        self.filename = filename
        self._log = []

    def __call__(self, trainer):
        import json
        self._log.append(dict(trainer.observation))
        path = os.path.join(trainer.out, self.filename)
        with open(path, 'w') as f:
            json.dump(self._log, f, indent=1, default=float)


class PrintReport(object):

    name = 'PrintReport'

    def __init__(self, keys):
        self.keys = keys

    def __call__(self, trainer):
        parts = []
        for k in self.keys:
            v = trainer.observation.get(k)
            parts.append('{}={}'.format(
                k, '{:.6g}'.format(v) if isinstance(v, float) else v))
        print('  '.join(parts), flush=True)
