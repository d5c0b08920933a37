# -*- coding: utf-8 -*-
"""Convergence oracles: fast deterministic runs that must learn.

The synthetic stand-in datasets use class-dependent Gaussian means, so a
linear model must exceed chance accuracy within a few rounds."""
import os

import pytest

import torch


def run_centered(fed_type='fedavg', rounds=4, workers=4, extra=None):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)
    argv = ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
            '--federated_type', fed_type, '--num_comms', str(rounds),
            '--online_client_rate', '1.0', '-b', '50', '--lr', '0.05',
            '--on_cuda', 'false', '-j', str(workers),
            '--checkpoint', '/tmp/ft_conv_ckpt', '--debug', 'false',
            '--manual_seed', '3'] + (extra or [])
    args = get_args(argv)
    args.num_workers = workers
    Clients = {}
    for i in range(workers):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    train_and_validate_federated_centered(Clients, Server)
    return Server


@pytest.mark.parametrize('fed_type', ['fedavg', 'scaffold'])
def test_centered_learns_synthetic_mnist(fed_type):
    server = run_centered(fed_type)
    acc = server.global_test_tracker['top1'].avg
    assert acc > 35.0, 'test top1 %.1f after training (chance=10)' % acc


def test_compressed_fedgate_learns():
    server = run_centered('fedgate', extra=['--compressed', 'true',
                                            '--compressed_ratio', '0.4'])
    acc = server.global_test_tracker['top1'].avg
    assert acc > 30.0, 'compressed top1 %.1f' % acc
