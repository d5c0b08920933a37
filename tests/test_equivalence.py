# -*- coding: utf-8 -*-
"""Centered == distributed fixed-seed equivalence oracle (the
reference's two execution modes implement the SAME math —
`/root/reference/main.py:29-42` vs `main_centered.py:31-43` — so the
single-process simulation and the 2-process gloo run must land on the same
server model).

Requires deterministic per-client loader order (`dataset._make_loader`
seeds each loader's generator from (manual_seed, client id)), identical
model init (seeded), and validation off in both modes (validation iterates
loaders and would advance their generators in centered mode only).
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

SEED = 7
ARGV_COMMON = [
    '-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
    '--num_comms', '2', '--online_client_rate', '1.0',
    '--federated_sync_type', 'local_step', '--local_step', '4',
    '-b', '20', '--lr', '0.1', '--in_momentum', 'true',
    '--on_cuda', 'false', '--debug', 'false',
    '--manual_seed', str(SEED)]


def _argv(fed_type, j, ckpt):
    argv = list(ARGV_COMMON) + ['--federated_type', fed_type, '-j', str(j),
                                '--checkpoint', ckpt]
    if fed_type == 'fedgate':
        argv += ['--compressed', 'false']
    if fed_type == 'fedadam':
        argv += ['--fedadam_beta', '0.9', '--fedadam_tau', '0.1']
    return argv


def _centered_worker(fed_type, q):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)

    args = get_args(_argv(fed_type, j=2, ckpt='/tmp/ft_eq_c_%s' % fed_type))
    # centered mode: -j doubles as the simulated-world size
    # (`main_centered.py:20`); force actual DataLoader workers to 0 so the
    # generator stream matches the workers=0 distributed loaders (with
    # persistent workers, re-iterating reuses the iterator and skips the
    # per-iterator base_seed draw that a fresh workers=0 iterator makes)
    import copy as _copy
    import fedtorch_amd.components.dataset as _ds
    _orig_ml = _ds._make_loader

    def _ml0(a, data, batch_size, shuffle, drop_last=False, tag=0):
        a = _copy.copy(a)
        a.num_workers = 0
        a.pin_memory = False
        return _orig_ml(a, data, batch_size, shuffle, drop_last, tag)
    _ds._make_loader = _ml0
    Clients = {}
    for i in range(2):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    Server.enable_grad(Clients[0].train_loader)
    train_and_validate_federated_centered(Clients, Server, validate=False)
    q.put(('server', Server.arena.clone_flat().numpy()))


def _dist_worker(rank, world, port, fed_type, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    dist.init_process_group('gloo', rank=rank, world_size=world)
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.federated import train_and_validate_federated

    args = get_args(_argv(fed_type, j=0, ckpt='/tmp/ft_eq_d_%s' % fed_type))
    client = Client(args, rank)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated(client, validate=False)
    if rank == 0:
        q.put(('server', client.arena.clone_flat().numpy()))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize('fed_type,port', [('fedavg', 29931),
                                           ('scaffold', 29933),
                                           ('fedgate', 29935),
                                           ('qsparse', 29937),
                                           ('fedadam', 29939)])
def test_centered_equals_distributed(fed_type, port):
    ctx = mp.get_context('spawn')

    qc = ctx.SimpleQueue()
    pc = ctx.Process(target=_centered_worker, args=(fed_type, qc))
    pc.start()

    qd = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dist_worker,
                         args=(r, 2, port, fed_type, qd))
             for r in range(2)]
    for p in procs:
        p.start()

    _, centered_flat = qc.get()
    _, dist_flat = qd.get()
    centered_flat = torch.from_numpy(centered_flat)
    dist_flat = torch.from_numpy(dist_flat)
    pc.join(timeout=300)
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    assert pc.exitcode == 0

    assert centered_flat.shape == dist_flat.shape
    # same math, different summation order (sequential accumulate vs
    # all-reduce): tight fp32 tolerance, not bitwise
    diff = (centered_flat - dist_flat).abs().max().item()
    assert torch.allclose(centered_flat, dist_flat, atol=5e-6, rtol=1e-5), \
        'max |centered - distributed| = %.3e (%s)' % (diff, fed_type)


def _centered_apfl_worker(q):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.apfl import (
        train_and_validate_apfl_centered)

    argv = list(ARGV_COMMON) + [
        '--federated_type', 'apfl', '--fed_personal', 'true',
        '--fed_personal_alpha', '0.5', '-j', '2',
        '--checkpoint', '/tmp/ft_eq_c_apfl']
    args = get_args(argv)
    import copy as _copy
    import fedtorch_amd.components.dataset as _ds
    _orig_ml = _ds._make_loader

    def _ml0(a, data, batch_size, shuffle, drop_last=False, tag=0):
        a = _copy.copy(a)
        a.num_workers = 0
        a.pin_memory = False
        return _orig_ml(a, data, batch_size, shuffle, drop_last, tag)
    _ds._make_loader = _ml0
    Clients = {}
    for i in range(2):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    Server.enable_grad(Clients[0].train_loader)
    train_and_validate_apfl_centered(Clients, Server, validate=False)
    q.put(('server', Server.arena.clone_flat().numpy(),
           Clients[0].arena_personal.clone_flat().numpy()))


def _dist_apfl_worker(rank, world, port, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    dist.init_process_group('gloo', rank=rank, world_size=world)
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.apfl import train_and_validate_federated_apfl

    argv = list(ARGV_COMMON) + [
        '--federated_type', 'apfl', '--fed_personal', 'true',
        '--fed_personal_alpha', '0.5', '-j', '0',
        '--checkpoint', '/tmp/ft_eq_d_apfl']
    args = get_args(argv)
    # the distributed apfl loop validates unconditionally (validation
    # advances loader generators): monkeypatch it out for the oracle
    import fedtorch_amd.trainings.apfl as am

    def _noval(*a, **k):
        return [0.0, 0.0, 0.0]
    am.do_validate = _noval
    client = Client(args, rank)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated_apfl(client)
    if rank == 0:
        q.put(('server', client.arena.clone_flat().numpy(),
               client.arena_personal.clone_flat().numpy()))
    dist.barrier()
    dist.destroy_process_group()


def test_apfl_centered_equals_distributed():
    """APFL (personalized) centered == 2-process distributed on fixed
    seeds: server model AND client 0's personal model."""
    ctx = mp.get_context('spawn')
    qc = ctx.SimpleQueue()
    pc = ctx.Process(target=_centered_apfl_worker, args=(qc,))
    pc.start()
    qd = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dist_apfl_worker, args=(r, 2, 29941, qd))
             for r in range(2)]
    for p in procs:
        p.start()
    _, c_srv, c_per = qc.get()
    _, d_srv, d_per = qd.get()
    pc.join(timeout=300)
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    assert pc.exitcode == 0
    import numpy as np
    ds = float(np.abs(c_srv - d_srv).max())
    dp = float(np.abs(c_per - d_per).max())
    assert ds < 5e-5, 'server diff %.3e' % ds
    assert dp < 5e-5, 'personal diff %.3e' % dp


def _centered_pp_worker(q):
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes.centered import ClientCentered, ServerCentered
    from fedtorch_amd.trainings.centered.main import (
        train_and_validate_federated_centered)
    argv = [a if a != '1.0' else '0.5' for a in ARGV_COMMON] + [
        '--federated_type', 'fedavg', '-j', '2',
        '--checkpoint', '/tmp/ft_eq_c_pp']
    args = get_args(argv)
    import copy as _copy
    import fedtorch_amd.components.dataset as _ds
    _orig = _ds._make_loader

    def _ml0(a, data, bs, sh, drop_last=False, tag=0):
        a = _copy.copy(a)
        a.num_workers = 0
        a.pin_memory = False
        return _orig(a, data, bs, sh, drop_last, tag)
    _ds._make_loader = _ml0
    Clients = {}
    for i in range(2):
        Clients[i] = ClientCentered(args, i) if i == 0 else \
            ClientCentered(args, i, Partitioner=Clients[0].Partitioner)
    Server = ServerCentered(Clients[0].args, Clients[0].model)
    Server.enable_grad(Clients[0].train_loader)
    train_and_validate_federated_centered(Clients, Server, validate=False)
    q.put(Server.arena.clone_flat().numpy())


def _dist_pp_worker(rank, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = '29943'
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    dist.init_process_group('gloo', rank=rank, world_size=2)
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.federated import train_and_validate_federated
    argv = [a if a != '1.0' else '0.5' for a in ARGV_COMMON] + [
        '--federated_type', 'fedavg', '-j', '0',
        '--checkpoint', '/tmp/ft_eq_d_pp']
    args = get_args(argv)
    client = Client(args, rank)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated(client, validate=False)
    if rank == 0:
        q.put(client.arena.clone_flat().numpy())
    dist.barrier()
    dist.destroy_process_group()


def test_centered_equals_distributed_partial_participation():
    """The oracle under online_client_rate < 1: the online-set sampling
    (np.random on rank 0, broadcast) must line up with the centered
    draw, and offline-round semantics must match across modes."""
    ctx = mp.get_context('spawn')
    qc = ctx.SimpleQueue()
    pc = ctx.Process(target=_centered_pp_worker, args=(qc,))
    pc.start()
    qd = ctx.SimpleQueue()
    procs = [ctx.Process(target=_dist_pp_worker, args=(r, qd))
             for r in range(2)]
    for p in procs:
        p.start()
    c = torch.from_numpy(qc.get())
    d = torch.from_numpy(qd.get())
    pc.join(timeout=300)
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    diff = (c - d).abs().max().item()
    assert torch.allclose(c, d, atol=5e-6, rtol=1e-5), \
        'partial-participation max diff %.3e' % diff
