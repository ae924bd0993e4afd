"""ObservationAggregator — cross-rank metric averaging.

Reference: chainermn/extensions/observation_aggregator.py [v6+] (~L1-90):
average a reporter observation (e.g. ``main/loss``) over ranks via
``allreduce_obj`` and republish it under ``aggregated_key``.
"""


class ObservationAggregator(object):

    name = 'ObservationAggregator'

    def __init__(self, comm, original_key, aggregated_key=None,
                 comm_trigger=(1, 'iteration')):
        self.comm = comm
        self.original_key = original_key
        self.aggregated_key = aggregated_key or original_key
        self.trigger = comm_trigger

    def __call__(self, trainer):
        value = trainer.observation.get(self.original_key)
        if value is None:
            value = 0.0
        total = self.comm.allreduce_obj(float(value))
        trainer.observation[self.aggregated_key] = total / self.comm.size
