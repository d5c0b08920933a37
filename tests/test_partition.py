# -*- coding: utf-8 -*-
"""Partitioner class-distribution properties."""
import types

import numpy as np
import torch

from fedtorch_amd.components.datasets.partition import (
    DataPartitioner, FederatedPartitioner)


class FakeData(object):
    def __init__(self, n=1000, classes=10, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.train_labels = torch.randint(0, classes, (n,), generator=g)

    def __len__(self):
        return len(self.train_labels)

    def __getitem__(self, i):
        return torch.zeros(1), self.train_labels[i]


def args_ns(n_nodes=4, **kw):
    ns = types.SimpleNamespace(
        is_distributed=False, data='cifar10', dirichlet=False,
        unbalanced=False, num_class_per_client=2, sensitive_feature=9, **kw)
    ns.graph = types.SimpleNamespace(rank=0, n_nodes=n_nodes,
                                     ranks=list(range(n_nodes)))
    return ns


def test_iid_equal_chunks():
    data = FakeData(1000)
    part = DataPartitioner(args_ns(), data, shuffle=True)
    sizes = [len(part.use(i)) for i in range(4)]
    assert sizes == [250, 250, 250, 250]
    all_idx = sum(part.partitions, [])
    assert len(set(all_idx)) == 1000


def test_noniid_classes_per_client():
    data = FakeData(2000)
    a = args_ns()
    part = FederatedPartitioner(a, data, shuffle=False)
    labels = data.train_labels
    for client in range(4):
        cls = labels[torch.tensor(part.partitions[client])].unique()
        # sorted-by-label slicing: a client sees far fewer classes than 10
        # (slices can straddle class boundaries, so allow ncpc*3)
        assert len(cls) <= a.num_class_per_client * 3


def test_dirichlet_split():
    np.random.seed(0)
    data = FakeData(4000)
    a = args_ns(n_nodes=8)
    a.dirichlet = True
    part = FederatedPartitioner(a, data, shuffle=False)
    assert len(part.partitions) == 8
    total = sum(len(p) for p in part.partitions)
    assert total <= 4000
    # heavy skew: most clients see few classes
    labels = data.train_labels
    n_few = sum(
        1 for p in part.partitions
        if len(p) and len(labels[torch.tensor(list(p))].unique()) <= 5)
    assert n_few >= 4


def test_unbalanced_sizes_differ():
    data = FakeData(2000)
    a = args_ns()
    a.unbalanced = True
    part = FederatedPartitioner(a, data, shuffle=False)
    sizes = [len(p) for p in part.partitions]
    assert len(set(sizes)) > 1
