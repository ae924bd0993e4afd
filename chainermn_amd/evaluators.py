"""Multi-node evaluator.

Reference: chainermn/evaluators.py (~L1-60): wrap an evaluator so
``evaluate()`` runs locally then ``allreduce_obj``-averages the result
dict — every rank reports identical metrics.
"""


def create_multi_node_evaluator(actual_evaluator, communicator):
    """Patch ``actual_evaluator.evaluate`` to average results over ranks."""
    actual_evaluate = actual_evaluator.evaluate

    def evaluate():
        local = actual_evaluate()
        summed = communicator.allreduce_obj(local)
        return {k: v / communicator.size for k, v in summed.items()}

    actual_evaluator.evaluate = evaluate
    return actual_evaluator
