from chainermn_amd.utils.bootstrap import ensure_initialized  # noqa: F401
from chainermn_amd.utils.bootstrap import get_topology  # noqa: F401
from chainermn_amd.utils.bootstrap import free_port  # noqa: F401
from chainermn_amd.utils.trace import trace_range, StepTimer  # noqa: F401
