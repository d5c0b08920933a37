# -*- coding: utf-8 -*-
"""MLP (parity with reference `fedtorch/components/models/nonconvex/mlp.py`).

BatchNorm runs with ``track_running_stats=False`` like the reference
(`mlp.py:25`) — under federation running stats would silently diverge per
client."""
import torch
import torch.nn as nn

_NUM_CLASSES = {'cifar10': 10, 'mnist': 10, 'fashion_mnist': 10, 'emnist': 10,
                'cifar100': 100, 'emnist_full': 62, 'adult': 2}


def _input_size(dataset):
    if 'cifar' in dataset:
        return 32 * 32 * 3
    if 'mnist' in dataset:
        return 28 * 28
    if dataset == 'adult':
        return 14
    raise NotImplementedError(dataset)


class MLP(nn.Module):
    def __init__(self, dataset, num_layers, hidden_size, drop_rate,
                 robust=False, track_running_stats=False):
        super().__init__()
        self.dataset = dataset
        self.num_layers = num_layers
        self.num_classes = _NUM_CLASSES[dataset]
        input_size = _input_size(dataset)
        if robust:
            self.noise = nn.Parameter(torch.randn(input_size) * 0.001)
        else:
            self.noise = None
        layers = []
        for i in range(num_layers):
            in_features = input_size if i == 0 else hidden_size
            layers.append(nn.Sequential(
                nn.Linear(in_features, hidden_size),
                nn.BatchNorm1d(hidden_size,
                               track_running_stats=track_running_stats),
                nn.ReLU(),
                nn.Dropout(p=drop_rate)))
        self.layers = nn.Sequential(*layers)
        self.fc = nn.Linear(hidden_size, self.num_classes, bias=False)

    def forward(self, x):
        out = x.reshape(x.size(0), -1)
        if self.noise is not None:
            out = out + self.noise
        out = self.layers(out)
        return self.fc(out)


def mlp(args):
    return MLP(dataset=args.data, num_layers=args.mlp_num_layers,
               hidden_size=args.mlp_hidden_size, drop_rate=args.drop_rate)


def robust_mlp(args):
    # reference robust_mlp keeps default BatchNorm (running stats on,
    # `robust_mlp.py:26`)
    return MLP(dataset=args.data, num_layers=args.mlp_num_layers,
               hidden_size=args.mlp_hidden_size, drop_rate=args.drop_rate,
               robust=True, track_running_stats=True)
