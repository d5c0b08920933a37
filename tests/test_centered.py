# -*- coding: utf-8 -*-
"""Centered-mode tests: aggregation math oracles + loop smoke."""
import types

import pytest
import torch
import torch.nn as nn

from fedtorch_amd.parallel.arena import Arena
from fedtorch_amd.components.optim.sgd import FusedSGD
from fedtorch_amd.aggregation import centered as C


class FakeNode(object):
    def __init__(self, seed, shift=0.0, args=None):
        torch.manual_seed(seed)
        self.model = nn.Linear(6, 3)
        self.arena = Arena(self.model)
        self.arena.flat.add_(shift)
        self.optimizer = FusedSGD(self.arena, lr=0.1)
        self.work = {}
        self.args = args


def fake_args(n_nodes=2, **kw):
    base = dict(quantized=False, compressed=False, federated_type='fedavg',
                lr_scale_at_sync=1.0, out_momentum=False, num_workers=n_nodes,
                compressed_ratio=0.5, online_client_rate=1.0,
                fedadam_beta=0.9, fedadam_tau=0.1, qffl_q=1.0)
    base.update(kw)
    ns = types.SimpleNamespace(**base)
    ns.graph = types.SimpleNamespace(rank=0, n_nodes=n_nodes,
                                     ranks=list(range(n_nodes)))
    return ns


def make_setup(fed_type='fedavg', n=2):
    args = fake_args(n, federated_type=fed_type)
    server = FakeNode(0, args=args)
    server.grad = server.arena.new_buffer()
    clients = {i: FakeNode(0, shift=float(i + 1), args=args)
               for i in range(n)}
    for c in clients.values():
        c.model_memory = c.arena.new_buffer()
        c.model_delta = c.arena.new_buffer()
        c.model_client_control = c.arena.new_buffer()
    server.model_server_control = server.arena.new_buffer()
    return args, server, clients


def test_centered_fedavg_mean():
    args, server, clients = make_setup()
    expected = torch.stack([c.arena.flat for c in clients.values()]).mean(0)
    C.fedavg_aggregation_centered(clients, server, [0, 1])
    assert torch.allclose(server.arena.flat, expected, atol=1e-6)


def test_centered_scaffold_runs_and_updates_controls():
    args, server, clients = make_setup('scaffold')
    before = server.arena.clone_flat()
    C.scaffold_aggregation_centered(clients, server, [0, 1], local_steps=5,
                                    lr=0.1)
    assert not torch.equal(server.arena.flat, before)
    # controls updated: c+ = (s - c)/(tau*lr), non-zero since clients shifted
    for c in clients.values():
        assert c.model_client_control.abs().sum() > 0
    assert torch.isfinite(server.model_server_control).all()


def test_centered_qsparse_error_feedback():
    args, server, clients = make_setup('qsparse')
    args.compressed = True
    C.qsparse_aggregation_centered(clients, server, [0, 1], local_steps=5,
                                   lr=0.1)
    # memory = (s - c) - agg for each client; finite and non-zero overall
    for c in clients.values():
        assert torch.isfinite(c.model_memory).all()
    assert torch.isfinite(server.arena.flat).all()


def test_centered_qffl_direction():
    args, server, clients = make_setup('qffl')
    for i, c in enumerate(clients.values()):
        c.full_loss = 1.0 + i
    before = server.arena.clone_flat()
    C.qffl_aggregation_centered(clients, server, [0, 1], lr=0.1)
    # moves toward clients (clients shifted positive => diff negative =>
    # p -= scale*agg increases)
    assert (server.arena.flat - before).mean() > 0
