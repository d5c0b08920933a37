# -*- coding: utf-8 -*-
"""WideResNet for CIFAR (parity with reference `nonconvex/wideresnet.py`).
Pre-activation relu(bn(x)) pairs use BNReLU so the fused NHWC BN kernels
fold the ReLU in on GPU (ops/batchnorm.py)."""
import torch.nn as nn
import torch.nn.functional as F

from fedtorch_amd.ops.batchnorm import BNReLU

_NUM_CLASSES = {'cifar10': 10, 'cifar100': 100, 'svhn': 10}


class _WideBlock(nn.Module):
    def __init__(self, in_planes, out_planes, stride, drop_rate=0.0):
        super().__init__()
        self.bn1 = BNReLU(in_planes)
        self.conv1 = nn.Conv2d(in_planes, out_planes, kernel_size=3,
                               stride=stride, padding=1, bias=False)
        self.bn2 = BNReLU(out_planes)
        self.conv2 = nn.Conv2d(out_planes, out_planes, kernel_size=3,
                               stride=1, padding=1, bias=False)
        self.drop_rate = drop_rate
        self.equal_io = in_planes == out_planes and stride == 1
        self.shortcut = None if self.equal_io else nn.Conv2d(
            in_planes, out_planes, kernel_size=1, stride=stride, bias=False)

    def forward(self, x):
        pre = self.bn1(x)
        out = self.conv1(pre)
        out = self.bn2(out)
        if self.drop_rate > 0:
            out = F.dropout(out, p=self.drop_rate, training=self.training)
        out = self.conv2(out)
        short = x if self.equal_io else self.shortcut(pre)
        return out + short


class WideResNet(nn.Module):
    def __init__(self, dataset, net_depth, widen_factor, drop_rate):
        super().__init__()
        assert (net_depth - 4) % 6 == 0, 'depth must be 6n+4'
        n = (net_depth - 4) // 6
        widths = [16, 16 * widen_factor, 32 * widen_factor, 64 * widen_factor]
        self.num_classes = _NUM_CLASSES[dataset]
        self.conv1 = nn.Conv2d(3, widths[0], kernel_size=3, padding=1,
                               bias=False)
        blocks = []
        in_planes = widths[0]
        for stage, (w, stride) in enumerate(zip(widths[1:], [1, 2, 2])):
            for i in range(n):
                blocks.append(_WideBlock(in_planes, w,
                                         stride if i == 0 else 1, drop_rate))
                in_planes = w
        self.blocks = nn.Sequential(*blocks)
        self.bn_final = BNReLU(widths[3])
        self.fc = nn.Linear(widths[3], self.num_classes)
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
        return self.fc(out)


def wideresnet(args):
    net_depth = int(args.arch.replace('wideresnet', '') or 28)
    return WideResNet(dataset=args.data, net_depth=net_depth,
                      widen_factor=args.wideresnet_widen_factor,
                      drop_rate=args.drop_rate)
