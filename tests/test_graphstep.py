# -*- coding: utf-8 -*-
"""hipGraph-captured local steps in the PARITY loops (--hip_graph):
CPU = flag is inert (stepper disables itself); GPU = fixed-seed
equivalence of the graph path vs the eager loop."""
import os

import pytest
import torch


def _run_single(hip_graph, seed=11, comms=2, on_cuda=False):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.federated import train_and_validate_federated
    argv = ['-d', 'mnist', '-a', 'cnn', '-f', 'true',
            '--federated_type', 'fedavg', '--num_comms', str(comms),
            '--online_client_rate', '1.0',
            '--federated_sync_type', 'local_step', '--local_step', '4',
            '-b', '20', '--lr', '0.1', '--in_momentum', 'true',
            '--on_cuda', 'true' if on_cuda else 'false',
            '--bf16', 'true' if on_cuda else 'false',
            '--hip_graph', 'true' if hip_graph else 'false',
            '--debug', 'false', '-j', '0', '--manual_seed', str(seed),
            '--checkpoint', '/tmp/ft_gs_%d' % int(hip_graph)]
    args = get_args(argv)
    client = Client(args, 0)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated(client, validate=False)
    return client.arena.clone_flat().float().cpu()


def test_hip_graph_flag_inert_on_cpu():
    a = _run_single(True)
    b = _run_single(False)
    assert torch.equal(a, b)


@pytest.mark.gpu
def test_hip_graph_matches_eager_on_gpu():
    """Same seeds, 2 rounds x 4 local steps: the captured-step trajectory
    must match the eager loop (state snapshot/restore around capture)."""
    a = _run_single(True, on_cuda=True)
    b = _run_single(False, on_cuda=True)
    diff = (a - b).abs().max().item()
    assert torch.allclose(a, b, atol=2e-3, rtol=1e-3), \
        'graph vs eager max diff %.2e' % diff


def _run_apfl(hip_graph, on_cuda=False, seed=19):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.apfl import train_and_validate_federated_apfl
    argv = ['-d', 'mnist', '-a', 'cnn', '-f', 'true',
            '--federated_type', 'apfl', '--fed_personal', 'true',
            '--fed_personal_alpha', '0.4', '--num_comms', '2',
            '--online_client_rate', '1.0',
            '--federated_sync_type', 'local_step', '--local_step', '4',
            '-b', '20', '--lr', '0.1', '--in_momentum', 'true',
            '--on_cuda', 'true' if on_cuda else 'false',
            '--bf16', 'true' if on_cuda else 'false',
            '--hip_graph', 'true' if hip_graph else 'false',
            '--debug', 'false', '-j', '0', '--manual_seed', str(seed),
            '--checkpoint', '/tmp/ft_gsa_%d' % int(hip_graph)]
    args = get_args(argv)
    client = Client(args, 0)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated_apfl(client)
    return (client.arena.clone_flat().float().cpu(),
            client.arena_personal.clone_flat().float().cpu())


def test_apfl_hip_graph_flag_inert_on_cpu():
    a = _run_apfl(True)
    b = _run_apfl(False)
    assert torch.equal(a[0], b[0]) and torch.equal(a[1], b[1])


@pytest.mark.gpu
def test_apfl_hip_graph_matches_eager_on_gpu():
    a = _run_apfl(True, on_cuda=True)
    b = _run_apfl(False, on_cuda=True)
    for i in range(2):
        d = (a[i] - b[i]).abs().max().item()
        assert torch.allclose(a[i], b[i], atol=3e-3, rtol=2e-3), \
            'model %d graph vs eager max diff %.2e' % (i, d)


def _run_loop(fed_type, hip_graph, on_cuda, drfa=False, seed=23):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    argv = ['-d', 'mnist', '-a', 'cnn', '-f', 'true',
            '--federated_type', fed_type, '--num_comms', '2',
            '--online_client_rate', '1.0',
            '--federated_sync_type', 'local_step', '--local_step', '4',
            '-b', '20', '--lr', '0.1', '--in_momentum', 'true',
            '--on_cuda', 'true' if on_cuda else 'false',
            '--bf16', 'true' if on_cuda else 'false',
            '--hip_graph', 'true' if hip_graph else 'false',
            '--debug', 'false', '-j', '0', '--manual_seed', str(seed),
            '--checkpoint', '/tmp/ft_gl_%s_%d' % (fed_type, int(hip_graph))]
    if drfa:
        argv += ['--federated_drfa', 'true']
    args = get_args(argv)
    client = Client(args, 0)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    if drfa:
        from fedtorch_amd.trainings.drfa import (
            train_and_validate_federated_drfa)
        # single-rank DRFA: k broadcast + lambda machinery degenerate
        import fedtorch_amd.trainings.drfa as dm
        torch.manual_seed(seed)  # fix the k draw across arms
        train_and_validate_federated_drfa(client)
    else:
        from fedtorch_amd.trainings.afl import (
            train_and_validate_federated_afl)
        train_and_validate_federated_afl(client)
    return client.arena.clone_flat().float().cpu()


@pytest.mark.gpu
def test_drfa_hip_graph_matches_eager_on_gpu():
    a = _run_loop('fedavg', True, True, drfa=True)
    b = _run_loop('fedavg', False, True, drfa=True)
    d = (a - b).abs().max().item()
    assert torch.allclose(a, b, atol=3e-3, rtol=2e-3), \
        'drfa graph vs eager diff %.2e' % d


@pytest.mark.gpu
def test_afl_hip_graph_matches_eager_on_gpu():
    a = _run_loop('afl', True, True)
    b = _run_loop('afl', False, True)
    d = (a - b).abs().max().item()
    assert torch.allclose(a, b, atol=3e-3, rtol=2e-3), \
        'afl graph vs eager diff %.2e' % d


def test_local_sgd_hip_graph_flag_inert_on_cpu():
    """plain distributed local SGD loop with --hip_graph on CPU: inert."""
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.local_sgd import train_and_validate

    def run(hg):
        argv = ['-d', 'mnist', '-a', 'cnn', '-f', 'false',
                '--stop_criteria', 'epoch', '--num_epochs', '1',
                '--local_step', '4', '-b', '20', '--lr', '0.1',
                '--in_momentum', 'true', '--on_cuda', 'false',
                '--hip_graph', 'true' if hg else 'false',
                '--debug', 'false', '-j', '0', '--manual_seed', '29',
                '--checkpoint', '/tmp/ft_ls_%d' % int(hg)]
        args = get_args(argv)
        client = Client(args, 0)
        client.initialize()
        client.initialize_dataset()
        client.load_local_dataset()
        client.gen_aux_models()
        train_and_validate(client)
        return client.arena.clone_flat().float().cpu()
    a = run(True)
    b = run(False)
    assert torch.equal(a, b)
