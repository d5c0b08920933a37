# -*- coding: utf-8 -*-
"""ResNet family (parity with reference `nonconvex/resnet.py`).

* ``ResNetCifar`` — 6n+2 sizes (ResNet-20 is the headline benchmark model,
  reference `resnet.py:209-257`): 3x3 stem to 16 channels, three stages
  16/32/64, Bottleneck blocks for size >= 44 like the reference.
* ``ResNetImageNet`` — 18/34/50/101/152 (reference `resnet.py:145-206`).

Written channels-last friendly (contiguous module graph, no functional
padding games) so MIOpen picks NHWC kernels under bf16 autocast on gfx950.
"""
import torch
import torch.nn as nn

from fedtorch_amd.ops.batchnorm import BNReLU, BNAddReLU

_NUM_CLASSES = {'cifar10': 10, 'cifar100': 100, 'svhn': 10, 'mnist': 10,
                'fashion_mnist': 10, 'emnist': 10, 'emnist_full': 62,
                'stl10': 10, 'imagenet': 1000}


def _num_classes(dataset):
    for key, n in _NUM_CLASSES.items():
        if key in dataset:
            return n
    raise NotImplementedError(dataset)


def conv3x3(in_planes, planes, stride=1):
    # NhwcConv3x3 = stock Conv2d whose channels_last bf16 wrw runs the MFMA
    # kernel (ops/conv3x3.py); every other configuration falls back inline.
    from fedtorch_amd.ops.conv3x3 import NhwcConv3x3
    return NhwcConv3x3(in_planes, planes, kernel_size=3, stride=stride,
                       padding=1, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = BNReLU(planes)  # BN+ReLU fused on GPU (ops/batchnorm.py)
        self.conv2 = conv3x3(planes, planes)
        # block tail relu(bn2(conv2) + identity): add+ReLU fold into the BN
        # kernels on GPU (ops/batchnorm.py BNAddReLU)
        self.bn2 = BNAddReLU(planes)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, kernel_size=1, bias=False)
        self.bn1 = BNReLU(planes)
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = BNReLU(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, kernel_size=1, bias=False)
        self.bn3 = BNAddReLU(planes * 4)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), identity)


class _ResNetBase(nn.Module):
    def _make_stage(self, block_fn, planes, block_num, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block_fn.expansion:
            from fedtorch_amd.ops.conv3x3 import NhwcConv1x1S2
            conv_cls = NhwcConv1x1S2 if stride == 2 else nn.Conv2d
            downsample = nn.Sequential(
                conv_cls(self.inplanes, planes * block_fn.expansion,
                         kernel_size=1, stride=stride, bias=False),
                nn.BatchNorm2d(planes * block_fn.expansion))
        layers = [block_fn(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block_fn.expansion
        for _ in range(1, block_num):
            layers.append(block_fn(self.inplanes, planes))
        return nn.Sequential(*layers)

    def _init_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out',
                                        nonlinearity='relu')
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)


class _Fp32Stem(nn.Module):
    """Run the 3-channel stem conv in fp32 even under bf16 autocast: MIOpen
    has no tuned bf16 NHWC wrw kernel for 3-input-channel convs and falls
    back to a naive kernel (profiles/r01_bench_notes.md)."""

    def __init__(self, conv):
        super().__init__()
        self.conv = conv

    def forward(self, x):
        with torch.autocast('cuda', enabled=False):
            return self.conv(x.float())


class ResNetCifar(_ResNetBase):
    def __init__(self, dataset, resnet_size):
        super().__init__()
        if resnet_size % 6 != 2:
            raise ValueError('resnet_size must be 6n + 2: %d' % resnet_size)
        block_num = (resnet_size - 2) // 6
        block_fn = Bottleneck if resnet_size >= 44 else BasicBlock
        self.num_classes = _num_classes(dataset)
        self.inplanes = 16
        self.conv1 = conv3x3(3, 16)
        self.bn1 = BNReLU(16)
        self.layer1 = self._make_stage(block_fn, 16, block_num)
        self.layer2 = self._make_stage(block_fn, 32, block_num, stride=2)
        self.layer3 = self._make_stage(block_fn, 64, block_num, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(64 * block_fn.expansion, self.num_classes)
        self._init_weights()

    def forward(self, x):
        x = self.bn1(self.conv1(x))
        x = self.layer3(self.layer2(self.layer1(x)))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


class ResNetImageNet(_ResNetBase):
    _PARAMS = {18: (BasicBlock, [2, 2, 2, 2]), 34: (BasicBlock, [3, 4, 6, 3]),
               50: (Bottleneck, [3, 4, 6, 3]), 101: (Bottleneck, [3, 4, 23, 3]),
               152: (Bottleneck, [3, 8, 36, 3])}

    def __init__(self, dataset, resnet_size):
        super().__init__()
        block_fn, block_nums = self._PARAMS[resnet_size]
        self.num_classes = _num_classes(dataset)
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, kernel_size=7, stride=2, padding=3,
                               bias=False)
        self.bn1 = BNReLU(64)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_stage(block_fn, 64, block_nums[0])
        self.layer2 = self._make_stage(block_fn, 128, block_nums[1], stride=2)
        self.layer3 = self._make_stage(block_fn, 256, block_nums[2], stride=2)
        self.layer4 = self._make_stage(block_fn, 512, block_nums[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block_fn.expansion, self.num_classes)
        self._init_weights()

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet(args):
    resnet_size = int(args.arch.replace('resnet', ''))
    if ('cifar' in args.data or 'svhn' in args.data or 'stl10' in args.data
            or 'downsampled_imagenet' in args.data):
        return ResNetCifar(dataset=args.data, resnet_size=resnet_size)
    if 'imagenet' in args.data:
        return ResNetImageNet(dataset=args.data, resnet_size=resnet_size)
    raise NotImplementedError(args.data)
