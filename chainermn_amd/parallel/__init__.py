from chainermn_amd.parallel.fused_dp import FusedDataParallelSGD  # noqa: F401
from chainermn_amd.parallel.fused_dp import FusedDataParallelLARS  # noqa: F401
from chainermn_amd.parallel.fused_dp import FusedDataParallelAdam  # noqa: F401
