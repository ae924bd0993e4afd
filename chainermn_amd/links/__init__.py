from chainermn_amd.links.batch_normalization import MultiNodeBatchNormalization  # noqa: F401
from chainermn_amd.links.create_mnbn_model import create_mnbn_model  # noqa: F401
from chainermn_amd.links.multi_node_chain_list import MultiNodeChainList  # noqa: F401
from chainermn_amd.links.n_step_rnn import create_multi_node_n_step_rnn  # noqa: F401
from chainermn_amd.links.n_step_rnn import MultiNodeNStepRNN  # noqa: F401
from chainermn_amd.links.fused_batch_norm import FusedBatchNorm2d  # noqa: F401
from chainermn_amd.links.fused_batch_norm import FusedBNReLU2d  # noqa: F401
from chainermn_amd.links.batch_normalization import MultiNodeBNReLU2d  # noqa: F401
from chainermn_amd.links.fused_pool import FusedMaxPool2d  # noqa: F401
from chainermn_amd.links.fused_conv import FusedConv1x1, FusedConv3x3  # noqa: F401
