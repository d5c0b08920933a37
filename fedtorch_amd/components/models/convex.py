# -*- coding: utf-8 -*-
"""Convex models (parity with reference `fedtorch/components/models/convex/`).

Per-dataset problem dims match `convex/logistic_regression.py:34-72` and
`convex/least_square.py:27-41`; weights are zero-initialized (`:76-80`).
The "robust" variants carry a learnable input-noise parameter trained by
gradient *ascent* with an L2-ball projection (handled in the training loops,
reference `comms/trainings/federated/main.py:131-141`).
"""
import torch
import torch.nn as nn

# dataset -> (num_features, num_classes) for classification-style convex models
_CLS_DIMS = {
    'epsilon': (2000, 2), 'url': (3231961, 2), 'rcv1': (47236, 2),
    'higgs': (28, 2), 'mnist': (784, 10), 'emnist': (784, 10),
    'emnist_full': (784, 62), 'cifar10': (3072, 10), 'cifar100': (3072, 100),
    'fashion_mnist': (784, 10), 'synthetic': (60, 10), 'adult': (14, 2),
}
_FLATTEN = ('mnist', 'cifar10', 'cifar100', 'fashion_mnist', 'emnist',
            'emnist_full')
# dataset -> num_features for regression (least squares)
_REG_DIMS = {'epsilon': 2000, 'url': 3231961, 'rcv1': 47236, 'MSD': 90}


class LogisticRegression(nn.Module):
    def __init__(self, dataset, robust=False):
        super().__init__()
        if dataset not in _CLS_DIMS:
            raise ValueError('unsupported dataset for convex model: %s' % dataset)
        self.dataset = dataset
        self.num_features, self.num_classes = _CLS_DIMS[dataset]
        if robust:
            self.noise = nn.Parameter(torch.randn(self.num_features) * 0.001)
        else:
            self.noise = None
        self.fc = nn.Linear(self.num_features, self.num_classes, bias=True)
        self.fc.weight.data.zero_()
        self.fc.bias.data.zero_()

    def forward(self, x):
        if self.dataset in _FLATTEN:
            x = x.reshape(-1, self.num_features)
        if self.noise is not None:
            x = x + self.noise
        return self.fc(x)


class LeastSquare(nn.Module):
    def __init__(self, dataset, robust=False):
        super().__init__()
        if dataset not in _REG_DIMS:
            raise ValueError('unsupported dataset for least squares: %s' % dataset)
        self.dataset = dataset
        self.num_features, self.num_classes = _REG_DIMS[dataset], 1
        if robust:
            self.noise = nn.Parameter(torch.randn(self.num_features) * 0.001)
        else:
            self.noise = None
        self.fc = nn.Linear(self.num_features, 1, bias=True)

    def forward(self, x):
        if self.noise is not None:
            x = x + self.noise
        return self.fc(x)


class LinearMAFL(nn.Module):
    """Two-factor linear model W@Z (reference
    `models/convex/least_square.py:43-67`): a server core `W` over a
    client-side projection `Z`, with the effective dense weight exposed as
    ``.weight``.  NOTE: in the reference this is dead code — no
    `--federated_type mafl` flag or training loop exists (`parameters.py`
    never defines `mafl_server_dim`); kept for API parity."""

    def __init__(self, in_features, middle_features, out_features=1):
        super().__init__()
        self.in_features = in_features if in_features else middle_features
        self.middle_features = middle_features
        self.out_features = out_features
        self.Z = nn.Linear(self.in_features, middle_features, bias=False)
        self.W = nn.Linear(middle_features, out_features, bias=True)
        self.core_params = self.W.parameters()
        self.extra_params = self.Z.parameters()

    @property
    def weight(self):
        return torch.matmul(self.W.weight, self.Z.weight)

    @property
    def bias(self):
        return self.W.bias

    def forward(self, x):
        return self.W(self.Z(x))


def logistic_regression(args):
    return LogisticRegression(dataset=args.data)


def robust_logistic_regression(args):
    return LogisticRegression(dataset=args.data, robust=True)


def least_square(args):
    if getattr(args, 'federated_type', None) == 'mafl':
        return LinearMAFL(in_features=getattr(args, 'input_dim', 0) or 0,
                          middle_features=getattr(args, 'mafl_server_dim',
                                                  10))
    return LeastSquare(dataset=args.data)


def robust_least_square(args):
    return LeastSquare(dataset=args.data, robust=True)
