# -*- coding: utf-8 -*-
"""End-to-end smoke matrix: one federated round for each dataset x model
family on CPU (single process, gloo)."""
import os

import pytest
import torch.distributed as dist

CASES = {
    'shakespeare_rnn': ['-d', 'shakespeare', '-a', 'rnn', '--iid_data',
                        'false', '-b', '10'],
    'epsilon_logreg': ['-d', 'epsilon', '-a', 'logistic_regression', '-b',
                       '32'],
    'msd_least_square': ['-d', 'MSD', '-a', 'least_square', '-b', '32'],
    'synthetic_logreg': ['-d', 'synthetic', '-a', 'logistic_regression',
                         '--iid_data', 'false', '-b', '25'],
    'cifar_cnn': ['-d', 'cifar10', '-a', 'cnn', '-b', '16'],
    'mnist_robust_mlp': ['-d', 'mnist', '-a', 'robust_mlp', '-b', '32'],
    'robust_logreg': ['-d', 'mnist', '-a', 'robust_logistic_regression',
                      '-b', '32'],
    'cifar_densenet': ['-d', 'cifar10', '-a', 'densenet', '-b', '8'],
    'cifar_resnet56': ['-d', 'cifar10', '-a', 'resnet56', '-b', '8'],
    'cifar_wideresnet': ['-d', 'cifar10', '-a', 'wideresnet', '-b', '8'],
    'mnist_dirichlet': ['-d', 'mnist', '-a', 'mlp', '--iid_data', 'false',
                        '--dirichlet', 'true', '-b', '16'],
    'mnist_unbalanced': ['-d', 'mnist', '-a', 'mlp', '--iid_data', 'false',
                         '--unbalanced', 'true', '--num_class_per_client',
                         '2', '-b', '16'],
}

_PORT = [29810]


@pytest.fixture(autouse=True)
def _dist_cleanup():
    yield
    if dist.is_initialized():
        dist.destroy_process_group()


@pytest.mark.parametrize('name', sorted(CASES))
def test_one_round(name, monkeypatch, tmp_path):
    monkeypatch.setenv('MASTER_ADDR', '127.0.0.1')
    _PORT[0] += 1
    monkeypatch.setenv('MASTER_PORT', str(_PORT[0]))
    monkeypatch.setenv('FEDTORCH_SYNTH_SIZE', '300')
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.main import main
    argv = CASES[name] + [
        '-f', 'true', '--federated_type', 'fedavg', '--num_comms', '1',
        '--online_client_rate', '1.0', '--lr', '0.05', '--on_cuda', 'false',
        '--dist_backend', 'gloo', '-j', '0',
        '--checkpoint', str(tmp_path), '--debug', 'false']
    main(get_args(argv))
