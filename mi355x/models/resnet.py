"""ResNet-18/50 on the mi355x op layer (BASELINE configs 2-5).

Structure follows the standard He et al. residual nets; the hot path is
our NHWC HIP kernels with BN+ReLU (and BN+add+ReLU at block exits) fused
into single kernels. `stem="cifar"` (3x3/1 conv, no maxpool — the standard
CIFAR ResNet stem) for 32x32 inputs; `stem="imagenet"` (7x7/2 + maxpool
3x3/2) for 224x224 (BASELINE config 4).
"""

from __future__ import annotations

import torch
from torch import nn

from .layers import (BatchNorm2d, Conv2d, Linear, MaxPool2d, _lazy_ok,
                     bn_conv_lazy, bn_then_conv, conv_bn, conv_bn_tap,
                     to_model_layout)
from mi355x import ops


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = Conv2d(in_planes, planes, 3, stride, 1, bias=False)
        self.bn1 = BatchNorm2d(planes, act="relu")
        self.conv2 = Conv2d(planes, planes, 3, 1, 1, bias=False)
        self.bn2 = BatchNorm2d(planes, act="relu")  # fused add+relu at exit
        self.downsample = downsample

    def forward(self, x):
        # lazy-BN: bn1's apply fuses into conv2's A-loads (the normalized
        # activation never materializes); conv1 carries the junction-grad
        # TAP and emits bn1's stats from its epilogue
        if _lazy_ok(self.bn1, self.conv2, x) and x.requires_grad:
            from mi355x.ops import functional as F_

            y1, tap, stats1 = F_.conv2d_tap_stats(
                x, self.conv1.weight, self.conv1.stride, self.conv1.padding)
            identity = (tap if self.downsample is None
                        else self.downsample(tap))
            y2, stats2 = bn_conv_lazy(self.bn1, self.conv2, y1, stats1,
                                      want_stats=True)
            return self.bn2(y2, residual=identity, stats=stats2)
        # the shortcut consumes conv1's TAP: the junction gradient fuses
        # into conv1's dgrad epilogue (layers.conv_bn_tap) instead of
        # autograd adding two full tensors at the block input
        out, tap = conv_bn_tap(self.conv1, self.bn1, x)
        identity = tap if self.downsample is None else self.downsample(tap)
        out = conv_bn(self.conv2, self.bn2, out, residual=identity)
        return out


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = Conv2d(in_planes, planes, 1, bias=False)
        self.bn1 = BatchNorm2d(planes, act="relu")
        self.conv2 = Conv2d(planes, planes, 3, stride, 1, bias=False)
        self.bn2 = BatchNorm2d(planes, act="relu")
        self.conv3 = Conv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = BatchNorm2d(planes * 4, act="relu")
        self.downsample = downsample

    def forward(self, x):
        # per-pair lazy dispatch: each internal BN fuses into its consumer
        # conv when eligible (bn_then_conv), else applies materialized —
        # one fused-stats chain either way
        if x.is_cuda and x.requires_grad and self.bn1.training and (
                _lazy_ok(self.bn1, self.conv2, x)
                or _lazy_ok(self.bn2, self.conv3, x)):
            from mi355x.ops import functional as F_

            y1, tap, stats1 = F_.conv2d_tap_stats(
                x, self.conv1.weight, self.conv1.stride, self.conv1.padding)
            identity = (tap if self.downsample is None
                        else self.downsample(tap))
            y2, stats2 = bn_then_conv(self.bn1, self.conv2, y1, stats1,
                                      want_stats=True)
            y3, stats3 = bn_then_conv(self.bn2, self.conv3, y2, stats2,
                                      want_stats=True)
            return self.bn3(y3, residual=identity, stats=stats3)
        out, tap = conv_bn_tap(self.conv1, self.bn1, x)
        identity = tap if self.downsample is None else self.downsample(tap)
        out = conv_bn(self.conv2, self.bn2, out)
        out = conv_bn(self.conv3, self.bn3, out, residual=identity)
        return out


class Downsample(nn.Module):
    def __init__(self, in_planes, out_planes, stride):
        super().__init__()
        self.conv = Conv2d(in_planes, out_planes, 1, stride, bias=False)
        self.bn = BatchNorm2d(out_planes)

    def forward(self, x):
        return conv_bn(self.conv, self.bn, x)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=10, stem="cifar"):
        super().__init__()
        self.in_planes = 64
        if stem == "cifar":
            self.conv1 = Conv2d(3, 64, 3, 1, 1, bias=False)
            self.maxpool = None
        else:
            self.conv1 = Conv2d(3, 64, 7, 2, 3, bias=False)
            self.maxpool = MaxPool2d(3, 2, 1)
        self.bn1 = BatchNorm2d(64, act="relu")
        self.layer1 = self._make_layer(block, 64, layers[0], 1)
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.fc = Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")

    def _make_layer(self, block, planes, blocks, stride):
        downsample = None
        if stride != 1 or self.in_planes != planes * block.expansion:
            downsample = Downsample(self.in_planes, planes * block.expansion,
                                    stride)
        layers = [block(self.in_planes, planes, stride, downsample)]
        self.in_planes = planes * block.expansion
        layers += [block(self.in_planes, planes) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = to_model_layout(x)  # NCHW -> NHWC, 16-bit on GPU
        x = conv_bn(self.conv1, self.bn1, x)
        if self.maxpool is not None:
            x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = ops.global_avg_pool(x)
        return self.fc(x)


def resnet18(num_classes=10, stem="cifar"):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, stem)


def resnet50(num_classes=1000, stem="imagenet"):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, stem)
