"""ResNet-50 — the benchmark workload (BASELINE.json configs 2/3/5).

Written from the architecture (He et al. 2015, bottleneck v1 with the
stride-on-3x3 refinement); the reference's equivalent was
examples/imagenet/models/resnet50.py.  ``norm_layer`` is injectable so
``create_mnbn_model`` / direct construction can use
MultiNodeBatchNormalization for the large-batch config.

channels_last memory format is the MI355X-friendly layout for MIOpen convs;
``bench.py`` converts the module, nothing here assumes NCHW strides.
"""

import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4
    # class-level switch for the residual-junction gradient fold
    # (tests compare fold-on vs fold-off on identical kernel stacks)
    _use_junction_fold = True
    # BN-backward reduce fused into the dgrad epilogues: numerically
    # verified (tests/test_gpu_kernels.py) but measured a NET LOSS even
    # after vectorizing the epilogue sweep (+62us at the GEMM's
    # 1-block/CU geometry vs the 29us standalone reduce at full
    # occupancy, layer3 shape) — ships disabled; see STATUS.md.
    _use_bnbwd_fold = False

    def __init__(self, in_ch, width, stride=1, downsample=None,
                 norm_layer=nn.BatchNorm2d, fused_ops=False, comm=None):
        super().__init__()
        out_ch = width * self.expansion
        self.fused_ops = fused_ops
        if fused_ops:
            bn_relu, bn_plain = _fused_bn_classes(comm)
            from chainermn_amd.links.fused_conv import (FusedConv1x1,
                                                        FusedConv3x3)
            conv1x1 = FusedConv1x1
            conv3x3 = FusedConv3x3
        else:
            bn_relu = bn_plain = norm_layer
            conv1x1 = (lambda ci, co: nn.Conv2d(ci, co, 1, bias=False))
            conv3x3 = (lambda ci, co, stride=1: nn.Conv2d(
                ci, co, 3, stride=stride, padding=1, bias=False))
        self.conv1 = conv1x1(in_ch, width)
        self.bn1 = bn_relu(width)
        self.conv2 = conv3x3(width, width, stride=stride)
        self.bn2 = bn_relu(width)
        self.conv3 = conv1x1(width, out_ch)
        self.bn3 = bn_plain(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        if self.fused_ops:
            # BN+ReLU and BN+Add+ReLU are single CDNA4 kernels; the conv
            # epilogues hand their (sum y | sum y^2) partials straight to
            # the following BN, skipping its standalone stats read pass.
            # Identity blocks also fork the input through _ResidualEntry:
            # in backward, conv1's dgrad GEMM accumulates the main-path
            # gradient into the identity gradient in its epilogue, so
            # the junction's separate grad-add never runs.
            junction = None
            x_main = x
            h1 = h2 = None
            if x.is_cuda:
                if self._use_bnbwd_fold:
                    # backward handoffs: bn1's backward reduce fuses
                    # into conv2's dgrad epilogue, bn2's into conv3's
                    h1, h2 = {}, {}
                if self.downsample is None and self._use_junction_fold:
                    from chainermn_amd.links.fused_conv import \
                        _ResidualEntry
                    junction = {}
                    x_main, identity = _ResidualEntry.apply(x, junction)
            y, st = self.conv1.forward_stats(x_main, junction=junction)
            out = self.bn1(y, pre_stats=st, bwd_handoff=h1)
            y, st = self.conv2.forward_stats(out, bwd_bn=h1)
            out = self.bn2(y, pre_stats=st, bwd_handoff=h2)
            if self.downsample is not None:
                identity = self.downsample(x)
            y, st = self.conv3.forward_stats(out, bwd_bn=h2)
            return self.bn3.forward_add_relu(y, identity, pre_stats=st,
                                             junction=junction)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        out += identity
        return self.relu(out)


def _fused_bn_classes(comm):
    """(bn_relu, bn_plain) classes: local fused BN, or multi-node fused BN
    when a communicator is given (global-batch statistics over RCCL)."""
    if comm is not None and comm.size > 1:
        from chainermn_amd.links.batch_normalization import (
            MultiNodeBatchNormalization, MultiNodeBNReLU2d)
        return (lambda c: MultiNodeBNReLU2d(c, comm),
                lambda c: MultiNodeBatchNormalization(c, comm))
    from chainermn_amd.links.fused_batch_norm import (FusedBatchNorm2d,
                                                      FusedBNReLU2d)
    return FusedBNReLU2d, FusedBatchNorm2d


class ResNet50(nn.Module):

    def __init__(self, num_classes=1000, norm_layer=nn.BatchNorm2d,
                 zero_init_residual=True, fused_ops=False, comm=None):
        super().__init__()
        self._comm = comm
        if fused_ops:
            self._stem_bn, norm_layer = _fused_bn_classes(comm)
        else:
            self._stem_bn = norm_layer
        self._norm = norm_layer
        self._fused_ops = fused_ops
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = self._stem_bn(64)
        self.relu = (nn.Identity() if fused_ops
                     else nn.ReLU(inplace=True))
        if fused_ops:
            from chainermn_amd.links.fused_pool import FusedMaxPool2d
            self.maxpool = FusedMaxPool2d(3, stride=2, padding=1)
        else:
            self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, 3, stride=1)
        self.layer2 = self._make_layer(128, 4, stride=2)
        self.layer3 = self._make_layer(256, 6, stride=2)
        self.layer4 = self._make_layer(512, 3, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        from chainermn_amd.links.fused_conv import (FusedConv1x1,
                                                    FusedConv3x3)
        for m in self.modules():
            if isinstance(m, (nn.Conv2d, FusedConv1x1, FusedConv3x3)):
                nn.init.kaiming_normal_(m.weight, mode='fan_out',
                                        nonlinearity='relu')
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck) and m.bn3.weight is not None:
                    nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width, blocks, stride):
        norm_layer = self._norm
        out_ch = width * Bottleneck.expansion
        downsample = None
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                norm_layer(out_ch))
        layers = [Bottleneck(self.in_ch, width, stride, downsample,
                             norm_layer, fused_ops=self._fused_ops,
                             comm=self._comm)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(out_ch, width, norm_layer=norm_layer,
                                     fused_ops=self._fused_ops,
                                     comm=self._comm))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(**kwargs):
    return ResNet50(**kwargs)
