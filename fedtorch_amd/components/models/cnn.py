# -*- coding: utf-8 -*-
"""LeNet-style CNN (parity with reference `nonconvex/cnn.py:9-63`)."""
import torch.nn as nn
import torch.nn.functional as F

_NUM_CLASSES = {'cifar10': 10, 'mnist': 10, 'fashion_mnist': 10, 'emnist': 10,
                'cifar100': 100, 'emnist_full': 62}


class CNN(nn.Module):
    def __init__(self, dataset):
        super().__init__()
        self.dataset = dataset
        self.num_classes = _NUM_CLASSES[dataset]
        self.num_channels = 3 if 'cifar' in dataset else 1
        self.rep_out_dim = 5 * 5 * 50 if 'cifar' in dataset else 4 * 4 * 50
        self.conv1 = nn.Conv2d(self.num_channels, 20, 5, 1)
        self.conv2 = nn.Conv2d(20, 50, 5, 1)
        self.fc1 = nn.Linear(self.rep_out_dim, 512)
        self.fc2 = nn.Linear(512, self.num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2, 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2, 2)
        x = x.reshape(-1, self.rep_out_dim)
        x = F.relu(self.fc1(x))
        return self.fc2(x)


def cnn(args):
    return CNN(args.data)
