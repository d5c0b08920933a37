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
