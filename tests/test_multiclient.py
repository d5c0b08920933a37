# -*- coding: utf-8 -*-
"""Virtual-client packing oracle: a rank packing C clients must produce the
SAME aggregate as C separate clients (weighted-diff partial sums are
linear)."""
import types

import torch
import torch.nn as nn

from fedtorch_amd.parallel.arena import Arena
from fedtorch_amd.parallel.multiclient import ClientPack
from fedtorch_amd.components.optim.sgd import FusedSGD


class FakeClient(object):
    def __init__(self, n_nodes=1, rank=0):
        torch.manual_seed(0)
        self.model = nn.Linear(8, 4)
        self.arena = Arena(self.model)
        self.optimizer = FusedSGD(self.arena, lr=0.1, in_momentum=0.9)
        self.args = types.SimpleNamespace(
            graph=types.SimpleNamespace(rank=rank, n_nodes=n_nodes,
                                        on_cuda=False))
        self.work = {}


def test_partial_equals_sum_of_diffs():
    client = FakeClient()
    pack = ClientPack(client, clients_per_rank=3)
    server = client.arena.clone_flat().add_(0.5)
    for j in range(3):
        pack.replicas[j].add_(float(j))
    w = [0.2, 0.0, 0.5]
    partial = pack.accumulate_partial(server, w)
    expected = sum(wj * (server - pack.replicas[j])
                   for j, wj in enumerate(w) if wj)
    assert torch.allclose(partial, expected, atol=1e-6)


def test_run_client_swaps_state():
    client = FakeClient()
    pack = ClientPack(client, clients_per_rank=2)
    server = client.arena.clone_flat()

    def fake_steps(loader):
        # a deterministic "local training": add 1 to all params via a fused
        # step with a constant gradient
        g = torch.ones_like(client.arena.grad)
        client.optimizer.step(apply_lr=True, apply_in_momentum=True,
                              apply_out_momentum=False, grad=g)
        return 1

    pack.run_client(0, server, fake_steps)
    pack.run_client(1, server, fake_steps)
    # both clients did the identical step from the same server state
    assert torch.allclose(pack.replicas[0], pack.replicas[1])
    assert not torch.allclose(pack.replicas[0], server)
    # momentum state is per client and initialized
    assert pack.mom_init == [True, True]
    assert torch.allclose(pack.in_mom[0], pack.in_mom[1])
    assert pack.in_mom[0].abs().sum() > 0


def test_pack_per_client_bn_buffers():
    """Each virtual client keeps its OWN BatchNorm running stats across the
    shared compute module (the reference's centered mode gives every
    client its own module; `parallel/multiclient.py` swaps buffer rows)."""
    import torch
    import torch.nn as nn
    from types import SimpleNamespace
    from fedtorch_amd.parallel.arena import Arena
    from fedtorch_amd.components.optim.sgd import FusedSGD

    torch.manual_seed(11)
    m = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.BatchNorm2d(4))
    arena = Arena(m)
    opt = FusedSGD(arena, lr=0.1, in_momentum=0.9)
    client = SimpleNamespace(
        arena=arena, optimizer=opt, model=m,
        args=SimpleNamespace(graph=SimpleNamespace(rank=0, n_nodes=2)))
    from fedtorch_amd.parallel.multiclient import ClientPack
    pack = ClientPack(client, 2)
    assert pack.bufs is not None and pack.bufs.shape[0] == 2

    server = arena.clone_flat()

    def steps_with_data(x):
        def run(loader):
            m(x)  # train-mode forward updates running stats
            return 1
        return run

    x0 = torch.randn(8, 3, 5, 5) * 3 + 7    # big mean/var
    x1 = torch.randn(8, 3, 5, 5) * 0.1      # small
    pack.train_loaders = [None, None]
    pack.run_client(0, server, steps_with_data(x0))
    pack.run_client(1, server, steps_with_data(x1))
    assert not torch.allclose(pack.bufs[0], pack.bufs[1])
    # client 0's stats reflect the large-mean batch, client 1's don't
    assert pack.bufs[0].abs().max() > pack.bufs[1].abs().max()
    # adopt_buffers: all rows equal the module's buffer arena
    arena.buf_flat.fill_(0.5)
    pack.adopt_buffers()
    assert torch.allclose(pack.bufs[0], pack.bufs[1])
    assert torch.allclose(pack.bufs[0],
                          torch.full_like(pack.bufs[0], 0.5))


def test_partial_buffers_weighted_by_online_count():
    """the BN partial must be weighted by each rank's
    online-client count, not a uniform per-rank mean.  Two packs with 2
    and 1 online clients (3 online total): the SUM of the two prescaled
    partials must equal the mean over the 3 online clients' stats, and a
    rank with zero online clients must contribute exact zeros (not its
    stale compute-module buffer)."""
    import torch.nn as nn
    from types import SimpleNamespace

    def make_pack(rank, C):
        torch.manual_seed(100 + rank)
        m = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.BatchNorm2d(4))
        arena = Arena(m)
        opt = FusedSGD(arena, lr=0.1, in_momentum=0.9)
        client = SimpleNamespace(
            arena=arena, optimizer=opt, model=m,
            args=SimpleNamespace(graph=SimpleNamespace(rank=rank,
                                                       n_nodes=2)))
        pack = ClientPack(client, C)
        for c in range(C):
            pack.bufs[c].normal_(float(rank * 10 + c), 1.0)
        return pack

    SimpleNamespace  # noqa: B018
    p0, p1 = make_pack(0, 2), make_pack(1, 2)
    total_online = 3
    p0.partial_buffers([0, 1], total_online)      # both local online
    p1.partial_buffers([1], total_online)         # one local online
    got = p0.base.arena.buf_flat + p1.base.arena.buf_flat
    want = (p0.bufs[0] + p0.bufs[1] + p1.bufs[1]) / 3.0
    assert torch.allclose(got, want, atol=1e-6)

    # zero online local clients -> exact zero contribution
    p1.base.arena.buf_flat.fill_(123.0)  # stale garbage
    p1.partial_buffers([], total_online)
    assert torch.equal(p1.base.arena.buf_flat,
                       torch.zeros_like(p1.base.arena.buf_flat))


def test_build_loaders_disjoint_partitions():
    """Every virtual client must get a DISJOINT chunk of the dataset:
    one shared partitioner per rank (building one per client gave each a
    different permutation — chunks could overlap)."""
    import os
    import types
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    args = get_args([
        '-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
        '--federated_type', 'fedavg', '--num_comms', '1',
        '--clients_per_rank', '4', '-b', '10', '--lr', '0.1',
        '--on_cuda', 'false', '-j', '0', '--manual_seed', '7',
        '--checkpoint', '/tmp/ft_dl', '--debug', 'false'])
    client = Client(args, 0)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    pack = ClientPack(client, 4)
    pack.build_loaders()
    types  # noqa: B018
    idx_sets = []
    for j in range(4):
        ds = pack.train_loaders[j].dataset
        base, idx = ds, None
        # unwrap Partition layers to base indices
        chain = []
        while hasattr(base, 'indices') and hasattr(base, 'data'):
            chain.append(list(base.indices))
            base = base.data
        # compose
        ids = chain[-1]
        for lvl in reversed(chain[:-1]):
            ids = [ids[i] for i in lvl]
        idx_sets.append(set(ids))
    union = set()
    for s_ in idx_sets:
        assert not (union & s_), 'virtual-client chunks overlap!'
        union |= s_
