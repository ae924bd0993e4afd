from chainermn_amd.training.trainer import Trainer  # noqa: F401
from chainermn_amd.training.trainer import Evaluator  # noqa: F401
from chainermn_amd.training.trainer import LogReport  # noqa: F401
from chainermn_amd.training.trainer import PrintReport  # noqa: F401
