# -*- coding: utf-8 -*-
"""DenseNet (BN+ReLU pairs as BNReLU: fused NHWC BN kernels fold the
ReLU on GPU) for CIFAR (parity with reference `nonconvex/densenet.py`).

Supports plain and BC mode (bottleneck + compression), growth rate and
compression per the reference factory (`densenet.py:200-208`).
"""
import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from fedtorch_amd.ops.batchnorm import BNReLU

_NUM_CLASSES = {'cifar10': 10, 'cifar100': 100, 'svhn': 10}


class _DenseLayer(nn.Module):
    def __init__(self, num_channels, growth_rate, drop_rate):
        super().__init__()
        self.bn1 = BNReLU(num_channels)
        self.conv1 = nn.Conv2d(num_channels, growth_rate, kernel_size=3,
                               padding=1, bias=False)
        self.drop_rate = drop_rate

    def forward(self, x):
        out = self.conv1(self.bn1(x))
        if self.drop_rate > 0:
            out = F.dropout(out, p=self.drop_rate, training=self.training)
        return torch.cat([x, out], 1)


class _BottleneckLayer(nn.Module):
    def __init__(self, num_channels, growth_rate, drop_rate):
        super().__init__()
        inter = 4 * growth_rate
        self.bn1 = BNReLU(num_channels)
        self.conv1 = nn.Conv2d(num_channels, inter, kernel_size=1, bias=False)
        self.bn2 = BNReLU(inter)
        self.conv2 = nn.Conv2d(inter, growth_rate, kernel_size=3, padding=1,
                               bias=False)
        self.drop_rate = drop_rate

    def forward(self, x):
        out = self.conv1(self.bn1(x))
        out = self.conv2(self.bn2(out))
        if self.drop_rate > 0:
            out = F.dropout(out, p=self.drop_rate, training=self.training)
        return torch.cat([x, out], 1)


class _Transition(nn.Module):
    def __init__(self, num_channels, num_out_channels, drop_rate):
        super().__init__()
        self.bn1 = BNReLU(num_channels)
        self.conv1 = nn.Conv2d(num_channels, num_out_channels, kernel_size=1,
                               bias=False)
        self.drop_rate = drop_rate

    def forward(self, x):
        out = self.conv1(self.bn1(x))
        if self.drop_rate > 0:
            out = F.dropout(out, p=self.drop_rate, training=self.training)
        return F.avg_pool2d(out, 2)


class DenseNet(nn.Module):
    def __init__(self, dataset, net_depth, growth_rate, bc_mode, compression,
                 drop_rate):
        super().__init__()
        self.num_classes = _NUM_CLASSES[dataset]
        layers_per_block = (net_depth - 4) // 3
        if bc_mode:
            layers_per_block //= 2
        num_channels = 2 * growth_rate
        self.conv1 = nn.Conv2d(3, num_channels, kernel_size=3, padding=1,
                               bias=False)
        blocks = []
        for i in range(3):
            layer_cls = _BottleneckLayer if bc_mode else _DenseLayer
            for _ in range(layers_per_block):
                blocks.append(layer_cls(num_channels, growth_rate, drop_rate))
                num_channels += growth_rate
            if i < 2:
                out_ch = int(math.floor(num_channels * compression)) \
                    if bc_mode else num_channels
                blocks.append(_Transition(num_channels, out_ch, drop_rate))
                num_channels = out_ch
        self.blocks = nn.Sequential(*blocks)
        self.bn_final = BNReLU(num_channels)
        self.classifier = nn.Linear(num_channels, self.num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out',
                                        nonlinearity='relu')
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def forward(self, x):
        out = self.blocks(self.conv1(x))
        out = self.bn_final(out)
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.classifier(out)


def densenet(args):
    net_depth = int(args.arch.replace('densenet', '') or 40)
    return DenseNet(dataset=args.data, net_depth=net_depth,
                    growth_rate=args.densenet_growth_rate,
                    bc_mode=args.densenet_bc_mode,
                    compression=args.densenet_compression,
                    drop_rate=args.drop_rate)
