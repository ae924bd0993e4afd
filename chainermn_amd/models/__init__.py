from chainermn_amd.models.mlp import MLP  # noqa: F401
from chainermn_amd.models.resnet import ResNet50  # noqa: F401
from chainermn_amd.models.resnet import resnet50  # noqa: F401
from chainermn_amd.models.seq2seq import Seq2Seq  # noqa: F401
