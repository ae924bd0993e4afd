"""Recursively replace BatchNorm modules with MultiNodeBatchNormalization.

Reference: chainermn/links/create_mnbn_model.py (~L1-60).
"""

import copy

import torch.nn as nn

from chainermn_amd.links.batch_normalization import (
    MultiNodeBatchNormalization, MultiNodeBNReLU2d)
from chainermn_amd.links.fused_batch_norm import (FusedBatchNorm2d,
                                                  FusedBNReLU2d)

_BN_TYPES = (nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d)


def create_mnbn_model(model, comm):
    """Return a copy of ``model`` with every BatchNormNd swapped for
    MultiNodeBatchNormalization (hyper-parameters and parameters copied)."""
    model = copy.deepcopy(model)
    _swap(model, comm)
    return model


def _swap(module, comm):
    for name, child in module.named_children():
        if isinstance(child, MultiNodeBatchNormalization):
            pass  # already multi-node
        elif isinstance(child, FusedBatchNorm2d):
            # fused local BN (incl. the +ReLU variant) -> the multi-node
            # class that keeps the same kernel fusions
            cls = (MultiNodeBNReLU2d if isinstance(child, FusedBNReLU2d)
                   else MultiNodeBatchNormalization)
            mnbn = cls(child.num_features, comm,
                       decay=1.0 - child.momentum, eps=child.eps,
                       use_gamma=child.weight is not None,
                       use_beta=child.bias is not None)
            if child.weight is not None:
                mnbn.weight.data.copy_(child.weight.data)
            if child.bias is not None:
                mnbn.bias.data.copy_(child.bias.data)
            if child.running_mean is not None:
                mnbn.running_mean.copy_(child.running_mean)
                mnbn.running_var.copy_(child.running_var)
            setattr(module, name, mnbn)
        elif isinstance(child, _BN_TYPES):
            mnbn = MultiNodeBatchNormalization(
                child.num_features, comm,
                decay=1.0 - child.momentum if child.momentum is not None
                else 0.9,
                eps=child.eps,
                use_gamma=child.weight is not None,
                use_beta=child.bias is not None)
            if child.weight is not None:
                mnbn.weight.data.copy_(child.weight.data)
            if child.bias is not None:
                mnbn.bias.data.copy_(child.bias.data)
            if child.running_mean is not None:
                mnbn.running_mean.copy_(child.running_mean)
                mnbn.running_var.copy_(child.running_var)
            setattr(module, name, mnbn)
        else:
            _swap(child, comm)
