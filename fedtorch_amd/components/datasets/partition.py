# -*- coding: utf-8 -*-
"""Data partitioners (parity with reference `components/datasets/partition.py`).

Kept semantics: rank-0 shuffles and broadcasts indices so every rank agrees
(`partition.py:25-33`); iid equal chunks (`:42-68`); growing-batch epochs
(`:71-104` — the reference's `is_dsitributed` typo that crashed the
distributed path is fixed here); non-iid federated splits (`:106-220`):
per-client pre-partitioned sets (synthetic/emnist/shakespeare), adult split
by sensitive feature, sort-by-label n-classes-per-client with optional
unbalanced sizes, and the Dirichlet(0.1) split (`--dirichlet`).
"""
import random

import numpy as np
import torch
import torch.distributed as dist


class Partition(object):
    """Index view over a dataset."""

    def __init__(self, data, indices):
        self.data = data
        self.indices = indices

    def __len__(self):
        return len(self.indices)

    def __getitem__(self, index):
        return self.data[self.indices[index]]


class Partitioner(object):
    def consistent_indices(self, indices, shuffle):
        """rank 0 shuffles; everyone takes rank 0's order (one broadcast on
        the WORLD group — the reference creates a fresh group per call,
        `partition.py:31`, which churns communicators for nothing)."""
        if self.args.graph.rank == 0 and shuffle:
            random.shuffle(indices)
        if dist.is_available() and dist.is_initialized():
            t = torch.tensor(indices, dtype=torch.int64)
            dist.broadcast(t, src=0)
            indices = t.tolist()
        return indices

    def check_indices(self, indices):
        if not (dist.is_available() and dist.is_initialized()):
            return
        t = torch.tensor(indices, dtype=torch.int64)
        ref = t.clone()
        dist.broadcast(ref, src=0)
        if not torch.equal(t, ref):
            raise ValueError('Data chunks in different devices are not the '
                             'same!')


class DataPartitioner(Partitioner):
    """iid equal chunks (reference `partition.py:42-68`)."""

    def __init__(self, args, data, shuffle, sizes=None):
        self.args = args
        self.data = data
        self.data_size = len(data)
        if sizes is None:
            sizes = [1.0 / args.graph.n_nodes] * args.graph.n_nodes
        self.partitions = []
        indices = list(range(self.data_size))
        if args.is_distributed:
            indices = self.consistent_indices(indices, shuffle)
        elif shuffle:
            random.shuffle(indices)
        from_index = 0
        for frac in sizes:
            to_index = from_index + int(frac * self.data_size)
            self.partitions.append(indices[from_index:to_index])
            from_index = to_index

    def use(self, partition_ind):
        return Partition(self.data, self.partitions[partition_ind])


class GrowingBatchPartitioner(Partitioner):
    """num_epochs copies of the index stream, per-rank slices per epoch
    (reference `partition.py:71-104`; its `is_dsitributed` typo fixed)."""

    def __init__(self, args, data, sizes=None):
        self.args = args
        self.data = data
        self.data_size_per_epoch = len(data)
        if sizes is None:
            sizes = [1.0 / args.graph.n_nodes] * args.graph.n_nodes
        self.partitions = []
        indices = []
        for _ in range(args.num_epochs):
            ind = list(range(self.data_size_per_epoch))
            if args.graph.rank == 0 and args.reshuffle_per_epoch:
                random.shuffle(ind)
            indices.extend(ind)
        if args.is_distributed:
            indices = self.consistent_indices(indices, False)
        from_index = 0
        for i in range(args.num_epochs):
            for ind, size in enumerate(sizes):
                to_index = from_index + int(size * self.data_size_per_epoch)
                if i == 0:
                    self.partitions.append(indices[from_index:to_index])
                else:
                    self.partitions[ind].extend(indices[from_index:to_index])
                from_index = to_index

    def use(self, partition_ind):
        return Partition(self.data, self.partitions[partition_ind])


class FederatedPartitioner(Partitioner):
    """Non-iid splits (reference `partition.py:106-220`)."""

    def __init__(self, args, data, shuffle, sizes=None):
        del sizes
        self.args = args
        self.data = data
        self.data_size = len(data)
        self.partitions = []

        if args.data in ('synthetic', 'emnist', 'emnist_full', 'shakespeare'):
            # per-client data was materialized per rank already.
            self.partitions = [[] for _ in range(args.graph.n_nodes)]
            self.partitions[args.graph.rank].extend(range(len(data)))
            return
        if args.data == 'adult':
            self._partition_adult()
            return
        self.labels = data.train_labels
        self.classes = self.labels.unique()
        if args.dirichlet:
            self._partition_dirichlet()
        else:
            self._partition_by_class(shuffle)

    def _partition_adult(self):
        args, data = self.args, self.data
        sens = data.features_name[args.sensitive_feature]
        groups = data.categories[sens]
        if args.graph.n_nodes % len(groups):
            raise ValueError('Number of nodes should be a multiple of the '
                             'number of sensitive groups')
        self.partitions = [[] for _ in range(args.graph.n_nodes)]
        per_group = args.graph.n_nodes // len(groups)
        train = data.train_data.numpy()
        for i, k in enumerate(groups):
            k_inds = np.where(
                train[:, args.sensitive_feature] == groups[k])[0].tolist()
            n_per_node = len(k_inds) // per_group
            from_index = 0
            for j in range(per_group):
                to_index = from_index + n_per_node \
                    if j != per_group - 1 else len(k_inds)
                self.partitions[i * per_group + j].extend(
                    k_inds[from_index:to_index])
                from_index = to_index

    def _partition_by_class(self, shuffle):
        """sort-by-label, num_class_per_client slices per client; optional
        unbalanced random sizes (reference `partition.py:145-183`)."""
        args = self.args
        n = args.graph.n_nodes
        ncpc = args.num_class_per_client
        if args.unbalanced:
            np.random.seed(1122)
            min_size = int(self.data_size / (len(self.classes) * n))
            slice_sizes = min_size * np.ones((ncpc, n), dtype=int)
            for i in range(ncpc):
                total_rem = int(self.data_size / ncpc) - min_size * n
                ind = np.sort(np.random.choice(
                    np.arange(0, total_rem), n - 1, replace=False))
                ind = np.concatenate([[0], ind, [total_rem]])
                slice_sizes[i, :] += ind[1:] - ind[:-1]
        else:
            slice_size = int(self.data_size / (n * ncpc))
            slice_sizes = np.full((ncpc, n), slice_size, dtype=int)

        label_array = np.asarray(self.labels)
        class_array = np.asarray(self.classes)
        indices = list(np.concatenate(
            [np.flatnonzero(label_array == c) for c in class_array]))
        if args.is_distributed:
            indices = self.consistent_indices(indices, shuffle=False)
        from_index = 0
        for n_class in range(ncpc):
            for client in range(n):
                to_index = from_index + slice_sizes[n_class, client]
                if n_class == 0:
                    self.partitions.append(list(indices[from_index:to_index]))
                else:
                    self.partitions[client].extend(indices[from_index:to_index])
                from_index = to_index

    def _partition_dirichlet(self):
        """Dirichlet(0.1) class mixture per client (reference
        `partition.py:184-203`, after arXiv:2003.13461)."""
        args = self.args
        n = args.graph.n_nodes
        client_data_size = int(self.data_size / n)
        label_array = np.asarray(self.labels)
        class_array = np.asarray(self.classes)
        class_ind_list = [np.flatnonzero(label_array == c)
                          for c in class_array]
        class_sample_size = [len(x) for x in class_ind_list]
        num_classes = len(class_array)
        probs = np.random.dirichlet(num_classes * [0.1 / num_classes], n)
        probs[probs * client_data_size < 10] = 0
        col = np.sum(probs, 0)
        # a class no client drew (column sum 0) gets no samples instead of
        # a 0/0 NaN -> invalid int cast
        probs = np.divide(probs * class_sample_size, col,
                          out=np.zeros_like(probs), where=col > 0)
        sample_sizes = probs.astype(int)
        ptr = np.zeros(num_classes, dtype=int)
        for client in range(n):
            parts = []
            for c in np.where(sample_sizes[client, :] > 0)[0]:
                to_index = ptr[c] + sample_sizes[client, c]
                parts.append(class_ind_list[c][ptr[c]:to_index])
                ptr[c] = to_index
            self.partitions.append(
                list(np.concatenate(parts)) if parts else [])

    def use(self, partition_ind):
        return Partition(self.data, self.partitions[partition_ind])
