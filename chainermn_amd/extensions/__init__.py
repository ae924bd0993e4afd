from chainermn_amd.extensions.checkpoint import create_multi_node_checkpointer  # noqa: F401
from chainermn_amd.extensions.allreduce_persistent import AllreducePersistent  # noqa: F401
from chainermn_amd.extensions.observation_aggregator import ObservationAggregator  # noqa: F401
from chainermn_amd.extensions.multi_node_snapshot import multi_node_snapshot  # noqa: F401
