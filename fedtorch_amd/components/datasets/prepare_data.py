# -*- coding: utf-8 -*-
"""Dataset dispatch (parity with reference `components/datasets/prepare_data.py`).

No network egress exists in this environment, so nothing downloads; loaders
read on-disk files when present and fall back to deterministic synthetic data
(see `sources.py`)."""
from fedtorch_amd.components.datasets import sources


def get_dataset(args, name, datasets_path, split='train'):
    if name in ('cifar10', 'cifar100', 'mnist', 'fashion_mnist', 'stl10'):
        return sources.get_vision_dataset(name, datasets_path, split)
    if name in ('emnist', 'emnist_full'):
        # federated EMNIST: real per-writer shards when materialized
        # (reference layout, `loader/federated_datasets.py:83-138`;
        # see `federated_shards.py`), else deterministic synthetic
        # stand-in shards seeded per client.
        import os
        from fedtorch_amd.components.datasets import federated_shards as fs
        root = os.path.join(datasets_path, name)
        if fs.emnist_shards_present(root):
            return fs.EMNISTShards(root, split=split,
                                   client_id=args.graph.rank)
        return sources.get_vision_dataset(
            name, datasets_path, split,
            seed=1234 + (args.graph.rank if split == 'train' else -1))
    if name == 'synthetic':
        return sources.get_synthetic_dataset(
            args, split, client_id=args.graph.rank)
    if name == 'shakespeare':
        return sources.get_shakespeare_dataset(
            args, split, client_id=args.graph.rank)
    if name == 'adult':
        return sources.get_adult_dataset(args, split)
    if name in ('epsilon', 'rcv1', 'higgs', 'MSD'):
        return sources.get_libsvm_dataset(args, name, split)
    raise NotImplementedError('dataset %s' % name)
