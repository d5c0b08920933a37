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


def test_centered_resnet_bn_stats_aggregated():
    """ResNet-20 federated training must LEARN (the reference never syncs
    BN running stats, so a BN-model server diverges in eval; we aggregate
    them — `aggregation/federated.py:aggregate_bn_buffers`)."""
    os.environ['FEDTORCH_SYNTH_SIZE'] = '512'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)
    args = get_args(['-d', 'cifar10', '-a', 'resnet20', '-f', 'true',
                     '--federated_type', 'fedavg', '--num_comms', '6',
                     '--online_client_rate', '1.0', '-b', '32', '--lr',
                     '0.05', '--in_momentum', 'true', '--on_cuda', 'false',
                     '--checkpoint', '/tmp/ft_conv_rn', '--debug', 'false',
                     '--manual_seed', '5'])
    args.num_workers = 2
    Clients = {0: ClientCentered(args, 0)}
    Clients[1] = ClientCentered(args, 1, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    train_and_validate_federated_centered(Clients, Server)
    acc = Server.global_test_tracker['top1'].avg
    loss = Server.global_test_tracker['losses'].avg
    assert loss < 2.5, 'server eval loss %.2f (BN stats broken?)' % loss
    assert acc > 25.0, 'test top1 %.1f' % acc


def test_centered_apfl_personal_learns():
    """APFL (personalized) centered loop learns past chance on the
    personal-blend validation (reference `centered/apfl.py`)."""
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.apfl import (
        train_and_validate_apfl_centered)
    argv = ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
            '--federated_type', 'apfl', '--fed_personal', 'true',
            '--fed_personal_alpha', '0.5', '--num_comms', '4',
            '--online_client_rate', '1.0', '-b', '50', '--lr', '0.05',
            '--on_cuda', 'false', '-j', '4',
            '--checkpoint', '/tmp/ft_conv_apfl', '--debug', 'false',
            '--manual_seed', '3']
    args = get_args(argv)
    args.num_workers = 4
    Clients = {}
    for i in range(4):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    train_and_validate_apfl_centered(Clients, Server)
    acc = Server.local_personal_val_tracker['top1'].avg
    assert acc > 30.0, acc


def test_centered_perfedme_learns():
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.perfedme import (
        train_and_validate_perfedme_centered)
    argv = ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
            '--federated_type', 'perfedme', '--num_comms', '4',
            '--online_client_rate', '1.0', '-b', '50', '--lr', '0.05',
            '--on_cuda', 'false', '-j', '4',
            '--checkpoint', '/tmp/ft_conv_pfm', '--debug', 'false',
            '--manual_seed', '3']
    args = get_args(argv)
    args.num_workers = 4
    Clients = {}
    for i in range(4):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    train_and_validate_perfedme_centered(Clients, Server)
    acc = Server.local_personal_val_tracker['top1'].avg
    assert acc > 25.0, acc


def _centered_clients(argv, workers=4):
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    args = get_args(argv)
    args.num_workers = workers
    Clients = {}
    for i in range(workers):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    return Clients, Server


def test_centered_afl_learns():
    """AFL centered: lambda-weighted aggregation still learns and the dual
    variable stays on the simplex (reference `centered/afl.py`)."""
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.trainings.centered.afl import (
        train_and_validate_afl_centered)
    Clients, Server = _centered_clients(
        ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
         '--federated_type', 'afl', '--num_comms', '4',
         '--online_client_rate', '1.0', '-b', '50', '--lr', '0.05',
         '--on_cuda', 'false', '-j', '4', '--checkpoint', '/tmp/ft_conv_afl',
         '--debug', 'false', '--manual_seed', '3'])
    train_and_validate_afl_centered(Clients, Server)
    lam = Server.lambda_vector
    assert abs(float(lam.sum()) - 1.0) < 1e-4
    assert float(lam.min()) >= 0.0
    acc = Server.global_test_tracker['top1'].avg
    assert acc > 30.0, acc


def test_centered_drfa_learns():
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.trainings.centered.drfa import (
        train_and_validate_drfa_centered)
    Clients, Server = _centered_clients(
        ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
         '--federated_type', 'fedavg', '--federated_drfa', 'true',
         '--num_comms', '4', '--online_client_rate', '0.75',
         '--local_step', '4', '--federated_sync_type', 'local_step',
         '-b', '50', '--lr', '0.05', '--on_cuda', 'false', '-j', '4',
         '--checkpoint', '/tmp/ft_conv_drfa', '--debug', 'false',
         '--manual_seed', '3'])
    train_and_validate_drfa_centered(Clients, Server)
    lam = Server.lambda_vector
    assert abs(float(lam.sum()) - 1.0) < 1e-4
    acc = Server.global_test_tracker['top1'].avg
    assert acc > 25.0, acc


def test_centered_qffl_learns():
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)
    Clients, Server = _centered_clients(
        ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
         '--federated_type', 'qffl', '--qffl_q', '1.0', '--num_comms', '4',
         '--online_client_rate', '1.0', '-b', '50', '--lr', '0.1',
         '--on_cuda', 'false', '-j', '4', '--checkpoint', '/tmp/ft_conv_qffl',
         '--debug', 'false', '--manual_seed', '3'])
    train_and_validate_federated_centered(Clients, Server)
    acc = Server.global_test_tracker['top1'].avg
    assert acc > 25.0, acc


def test_centered_perfedavg_learns():
    os.environ['FEDTORCH_SYNTH_SIZE'] = '600'
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)
    Clients, Server = _centered_clients(
        ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
         '--federated_type', 'perfedavg', '--num_comms', '4',
         '--online_client_rate', '1.0', '-b', '50', '--lr', '0.05',
         '--on_cuda', 'false', '-j', '4',
         '--checkpoint', '/tmp/ft_conv_pfa', '--debug', 'false',
         '--manual_seed', '3'])
    train_and_validate_federated_centered(Clients, Server)
    acc = Server.local_personal_val_tracker['top1'].avg
    assert acc > 25.0, acc
