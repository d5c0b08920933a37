# -*- coding: utf-8 -*-
"""Dataset sources — self-contained loaders (no torchvision / h5py in this
environment, and no network egress).

Every loader reads standard on-disk formats when present under
``args.data_dir`` and otherwise falls back to a DETERMINISTIC synthetic
dataset of the right shape (loudly logged).  The reference downloads via
torchvision / TFF HDF5 (`components/datasets/prepare_data.py:15-163`,
`loader/federated_datasets.py`); download is impossible here, so synthetic
fallback is what keeps every config runnable (and is what the benchmarks use
— BASELINE.md mandates synthetic data).

The ``Synthetic(alpha, beta)`` federated dataset reproduces the reference
generator math exactly (`loader/federated_datasets.py:204-296`): per-client
lognormal sample counts, x ~ N(B_k 1, Sigma) with Sigma_ii = (i+1)^-1.2,
y = argmax softmax(x w + noise).
"""
import gzip
import os
import pickle

import numpy as np
import torch

from fedtorch_amd.logs.logging import log


def _synth_size(default):
    """Allow tests to shrink fallback datasets: FEDTORCH_SYNTH_SIZE env."""
    v = os.environ.get('FEDTORCH_SYNTH_SIZE')
    return int(v) if v else default


class ArrayDataset(torch.utils.data.Dataset):
    """(x, y) tensor dataset with the attributes the partitioners expect."""

    def __init__(self, x, y, classes=None, transform=None):
        assert len(x) == len(y)
        self.x = x
        self.y = y
        self.transform = transform
        self.train_labels = y if y.dtype == torch.long else None
        if classes is not None:
            self.classes = classes

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        xi = self.x[i]
        if self.transform is not None:
            xi = self.transform(xi)
        return xi, self.y[i]


# --------------------------------------------------------------------------
# vision: CIFAR / MNIST-family raw parsers + synthetic fallback
# --------------------------------------------------------------------------

_VISION_SHAPES = {
    'cifar10': ((3, 32, 32), 10, 50000, 10000),
    'cifar100': ((3, 32, 32), 100, 50000, 10000),
    'mnist': ((1, 28, 28), 10, 60000, 10000),
    'fashion_mnist': ((1, 28, 28), 10, 60000, 10000),
    'emnist': ((1, 28, 28), 10, 60000, 10000),
    'emnist_full': ((1, 28, 28), 62, 60000, 10000),
    'stl10': ((3, 96, 96), 10, 5000, 8000),
}

_CIFAR_MEAN = torch.tensor([0.4914, 0.4822, 0.4465]).view(3, 1, 1)
_CIFAR_STD = torch.tensor([0.2470, 0.2435, 0.2616]).view(3, 1, 1)
_MNIST_MEAN, _MNIST_STD = 0.1307, 0.3081


def _cifar_train_transform(x):
    """random crop(pad 4) + horizontal flip, in plain torch ops."""
    x = torch.nn.functional.pad(x, (4, 4, 4, 4), mode='reflect'
                                ) if x.dim() == 3 else x
    i = torch.randint(0, 9, (2,))
    x = x[:, i[0]:i[0] + 32, i[1]:i[1] + 32]
    if torch.rand(()) < 0.5:
        x = torch.flip(x, dims=(2,))
    return x


def _read_idx(path):
    opener = gzip.open if path.endswith('.gz') else open
    with opener(path, 'rb') as f:
        data = f.read()
    magic = int.from_bytes(data[2:3], 'big')
    ndim = data[3]
    dims = [int.from_bytes(data[4 + 4 * i:8 + 4 * i], 'big')
            for i in range(ndim)]
    arr = np.frombuffer(data, dtype=np.uint8, offset=4 + 4 * ndim)
    del magic
    return arr.reshape(dims)


def _load_mnist_raw(root, train):
    prefix = 'train' if train else 't10k'
    for ext in ('', '.gz'):
        ip = os.path.join(root, '{}-images-idx3-ubyte{}'.format(prefix, ext))
        lp = os.path.join(root, '{}-labels-idx1-ubyte{}'.format(prefix, ext))
        if os.path.exists(ip) and os.path.exists(lp):
            images = _read_idx(ip).astype(np.float32) / 255.0
            labels = _read_idx(lp).astype(np.int64)
            return torch.from_numpy(images).unsqueeze(1), \
                torch.from_numpy(labels)
    return None


def _load_cifar_raw(root, name, train):
    if name == 'cifar10':
        base = os.path.join(root, 'cifar-10-batches-py')
        files = ['data_batch_%d' % i for i in range(1, 6)] if train \
            else ['test_batch']
        label_key = b'labels'
    else:
        base = os.path.join(root, 'cifar-100-python')
        files = ['train'] if train else ['test']
        label_key = b'fine_labels'
    if not os.path.isdir(base):
        return None
    xs, ys = [], []
    for fn in files:
        p = os.path.join(base, fn)
        if not os.path.exists(p):
            return None
        with open(p, 'rb') as f:
            d = pickle.load(f, encoding='bytes')
        xs.append(d[b'data'])
        ys.extend(d[label_key])
    x = np.concatenate(xs).reshape(-1, 3, 32, 32).astype(np.float32) / 255.0
    return torch.from_numpy(x), torch.tensor(ys, dtype=torch.long)


def get_vision_dataset(name, root, split, seed=1234):
    """CIFAR/MNIST-family; real files if present, synthetic fallback else."""
    shape, num_classes, n_train, n_test = _VISION_SHAPES[name]
    train = split == 'train'
    data = None
    if 'cifar' in name:
        data = _load_cifar_raw(root, name, train)
    elif name in ('mnist', 'fashion_mnist'):
        sub = os.path.join(root, 'FashionMNIST' if name == 'fashion_mnist'
                           else 'MNIST', 'raw')
        data = _load_mnist_raw(sub if os.path.isdir(sub) else root, train)
    if data is None:
        n = _synth_size(n_train if train else n_test)
        log('WARNING: no on-disk files for dataset {} under {} — generating '
            'a deterministic SYNTHETIC stand-in ({} samples, seed {}).'
            .format(name, root, n, seed), debug=True)
        # LEARNABLE synthetic stand-in: class-dependent Gaussian means, so
        # convergence / rounds-to-accuracy stay meaningful without real
        # data. Train and test share templates (same base seed).
        gt = torch.Generator().manual_seed(seed * 1000 + 7)
        templates = torch.randn((num_classes,) + shape, generator=gt) * 0.4
        g = torch.Generator().manual_seed(seed + (0 if train else 1))
        y = torch.randint(0, num_classes, (n,), generator=g)
        x = (torch.rand((n,) + shape, generator=g) * 0.8 +
             templates[y]).clamp_(0, 1)
    else:
        x, y = data
    # normalize
    if 'cifar' in name:
        x = (x - _CIFAR_MEAN) / _CIFAR_STD
        transform = _cifar_train_transform if train else None
    else:
        x = (x - _MNIST_MEAN) / _MNIST_STD
        transform = None
    return ArrayDataset(x, y, classes=torch.arange(num_classes),
                        transform=transform)


# --------------------------------------------------------------------------
# Synthetic(alpha, beta) federated dataset (reference generator math)
# --------------------------------------------------------------------------

def _softmax_np(z, axis=1):
    z = z - z.max(axis=axis, keepdims=True)
    e = np.exp(z)
    return e / e.sum(axis=axis, keepdims=True)


class SyntheticAllTasks(object):
    """Generates ALL clients' tasks deterministically in one pass (so every
    rank agrees without files), reference `federated_datasets.py:204-296`."""

    def __init__(self, alpha, beta, num_tasks, seed=931231, num_dim=60,
                 num_classes=10, min_num_samples=500, max_num_samples=1000,
                 regression=False, test_ratio=0.2):
        rng = np.random.RandomState(seed)
        self.regression = regression
        if regression:
            num_classes = 1
        Sigma = np.diag([(i + 1) ** (-1.2) for i in range(num_dim)])
        num_samples = rng.lognormal(3, 2, num_tasks).astype(int)
        num_samples = [min(s + min_num_samples, max_num_samples)
                       for s in num_samples]
        self.train, test_x, test_y = [], [], []
        for s in num_samples:
            B = rng.normal(loc=0.0, scale=beta)
            loc = rng.normal(loc=B, scale=1.0, size=num_dim)
            x = np.ones((s, num_dim + 1))
            x[:, 1:] = rng.multivariate_normal(mean=loc, cov=Sigma, size=s)
            wloc = rng.normal(loc=0, scale=alpha)
            w = rng.normal(loc=wloc, scale=1, size=(num_dim + 1, num_classes))
            out = x @ w + rng.normal(loc=wloc, scale=0.1, size=(s, num_classes))
            if regression:
                y = np.squeeze(out)
            else:
                y = np.argmax(_softmax_np(out), axis=1)
            x = x[:, 1:]
            shuffle = rng.permutation(s)
            cut = int(s * (1 - test_ratio))
            tr, te = shuffle[:cut], shuffle[cut:]
            ydt = np.float32 if regression else np.int64
            self.train.append((
                torch.from_numpy(x[tr].astype(np.float32)),
                torch.from_numpy(y[tr].astype(ydt))))
            test_x.append(x[te].astype(np.float32))
            test_y.append(y[te].astype(ydt))
        self.test = (torch.from_numpy(np.concatenate(test_x)),
                     torch.from_numpy(np.concatenate(test_y)))


_SYNTH_CACHE = {}


def get_synthetic_dataset(args, split, client_id):
    key = (args.synthetic_alpha, args.synthetic_beta, args.graph.n_nodes,
           'least_square' in args.arch)
    if key not in _SYNTH_CACHE:
        _SYNTH_CACHE[key] = SyntheticAllTasks(
            alpha=args.synthetic_alpha, beta=args.synthetic_beta,
            num_tasks=args.graph.n_nodes,
            regression='least_square' in args.arch)
    gen = _SYNTH_CACHE[key]
    if split == 'train':
        x, y = gen.train[client_id]
    else:
        x, y = gen.test
    classes = None if gen.regression else torch.arange(10)
    return ArrayDataset(x, y, classes=classes)


# --------------------------------------------------------------------------
# Shakespeare char stream (per-client) with synthetic fallback
# --------------------------------------------------------------------------

class CharWindows(torch.utils.data.Dataset):
    """(seq_len+1)-char windows -> (x = w[:-1], y = w[1:]) like the reference
    (`federated_datasets.py:364-368`)."""

    def __init__(self, codes, seq_len):
        n = (len(codes) - 1) // seq_len
        codes = codes[:n * seq_len + 1]
        self.x = codes[:-1].reshape(n, seq_len)
        self.y = codes[1:].reshape(n, seq_len)
        self.train_labels = None

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def get_shakespeare_dataset(args, split, client_id):
    root = os.path.join(args.data_dir, 'shakespeare')
    # real per-role shards (reference layout,
    # `loader/federated_datasets.py:390-472`) take priority
    from fedtorch_amd.components.datasets import federated_shards as fs
    if fs.shakespeare_shards_present(root):
        return fs.ShakespeareShards(root, split=split, client_id=client_id,
                                    seq_len=args.rnn_seq_len)
    pt = os.path.join(root, 'Client_{}.pt'.format(client_id)
                      if split == 'train' else 'Test.pt')
    if os.path.exists(pt):
        codes = torch.load(pt)
    else:
        n_chars = _synth_size(40000)
        log('WARNING: shakespeare files not found under {} — synthetic '
            'char stream ({} chars).'.format(root, n_chars), debug=True)
        g = torch.Generator().manual_seed(
            1000 + (client_id if split == 'train' else -1))
        codes = torch.randint(0, args.vocab_size, (n_chars,), generator=g)
    return CharWindows(codes.long(), args.rnn_seq_len)


# --------------------------------------------------------------------------
# UCI Adult with sensitive-feature categories; synthetic fallback
# --------------------------------------------------------------------------

_ADULT_CAT_COLS = {1, 3, 5, 6, 7, 8, 9, 13}  # categorical columns


class AdultDataset(ArrayDataset):
    def __init__(self, x, y, categories, features_name, split):
        super().__init__(x, y, classes=torch.arange(2))
        self.categories = categories
        self.features_name = features_name
        if split == 'train':
            self.train_data = x

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def get_adult_dataset(args, split, seed=77):
    path = os.path.join(args.data_dir, 'adult',
                        'adult.data' if split == 'train' else 'adult.test')
    features_name = ['age', 'workclass', 'fnlwgt', 'education',
                     'education-num', 'marital-status', 'occupation',
                     'relationship', 'race', 'sex', 'capital-gain',
                     'capital-loss', 'hours-per-week', 'native-country']
    if os.path.exists(path):
        rows = []
        with open(path) as f:
            for line in f:
                parts = [p.strip() for p in line.strip().rstrip('.').split(',')]
                if len(parts) == 15:
                    rows.append(parts)
        categories = {}
        for c in _ADULT_CAT_COLS:
            vals = sorted({r[c] for r in rows})
            categories[features_name[c]] = {v: i for i, v in enumerate(vals)}
        x = np.zeros((len(rows), 14), dtype=np.float32)
        y = np.zeros(len(rows), dtype=np.int64)
        for i, r in enumerate(rows):
            for c in range(14):
                x[i, c] = categories[features_name[c]][r[c]] \
                    if c in _ADULT_CAT_COLS else float(r[c])
            y[i] = 1 if '>50K' in r[14] else 0
        # normalize continuous columns
        for c in range(14):
            if c not in _ADULT_CAT_COLS:
                mu, sd = x[:, c].mean(), x[:, c].std() + 1e-8
                x[:, c] = (x[:, c] - mu) / sd
    else:
        n = _synth_size(30000 if split == 'train' else 15000)
        log('WARNING: adult data not found at {} — synthetic stand-in '
            '({} rows).'.format(path, n), debug=True)
        rng = np.random.RandomState(seed + (0 if split == 'train' else 1))
        x = rng.randn(n, 14).astype(np.float32)
        # sensitive feature becomes a 2-category column
        x[:, args.sensitive_feature] = rng.randint(0, 2, n)
        y = (rng.rand(n) > 0.5).astype(np.int64)
        categories = {features_name[c]: ({'a': 0, 'b': 1}
                      if c == args.sensitive_feature else {'a': 0})
                      for c in _ADULT_CAT_COLS | {args.sensitive_feature}}
    return AdultDataset(torch.from_numpy(np.asarray(x)),
                        torch.from_numpy(np.asarray(y)),
                        categories, features_name, split)


# --------------------------------------------------------------------------
# LibSVM-format datasets (epsilon / rcv1 / higgs / MSD); synthetic fallback
# --------------------------------------------------------------------------

_LIBSVM_DIMS = {'epsilon': 2000, 'rcv1': 47236, 'higgs': 28, 'MSD': 90}


def get_libsvm_dataset(args, name, split, seed=55):
    dim = _LIBSVM_DIMS[name]
    regression = name == 'MSD'
    path = os.path.join(args.data_dir, name,
                        '{}.{}'.format(name, 'train' if split == 'train'
                                       else 'test'))
    if os.path.exists(path):
        xs, ys = [], []
        with open(path) as f:
            for line in f:
                parts = line.split()
                if not parts:
                    continue
                label = float(parts[0])
                row = np.zeros(dim, dtype=np.float32)
                for kv in parts[1:]:
                    k, v = kv.split(':')
                    row[int(k) - 1] = float(v)
                xs.append(row)
                ys.append(label)
        x = torch.from_numpy(np.stack(xs))
        yarr = np.asarray(ys)
        if regression:
            y = torch.from_numpy(yarr.astype(np.float32))
        else:
            y = torch.from_numpy((yarr > 0).astype(np.int64))
    else:
        n = _synth_size(20000 if split == 'train' else 4000)
        log('WARNING: {} data not found at {} — synthetic stand-in '
            '({} rows).'.format(name, path, n), debug=True)
        rng = np.random.RandomState(seed + (0 if split == 'train' else 1))
        x = torch.from_numpy(rng.randn(n, dim).astype(np.float32))
        if regression:
            y = torch.from_numpy(rng.randn(n).astype(np.float32))
        else:
            y = torch.from_numpy((rng.rand(n) > 0.5).astype(np.int64))
    classes = None if regression else torch.arange(2)
    return ArrayDataset(x, y, classes=classes)
