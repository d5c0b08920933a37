# -*- coding: utf-8 -*-
"""2-process gloo integration tests (BASELINE config 1: plumbing on CPU).

Checks the strongest invariants of the sync semantics:
* after a round, all ranks hold the SAME model (bitwise);
* with full participation and equal weights, the aggregate equals the
  mean of the client models (lr_scale_at_sync=1).
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port, fed_type, q):
    try:
        _worker_inner(rank, world, port, fed_type, q)
    except Exception as e:  # noqa: BLE001
        q.put((rank, False, float('nan')))
        raise


def _worker_inner(rank, world, port, fed_type, q):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
    dist.init_process_group('gloo', rank=rank, world_size=world)
    from fedtorch_amd.parameters import get_args
    from fedtorch_amd.nodes import Client
    from fedtorch_amd.trainings.federated import train_and_validate_federated

    argv = ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
            '--federated_type', fed_type, '--num_comms', '2',
            '--online_client_rate', '1.0', '--num_epochs_per_comm', '1',
            '-b', '25', '--lr', '0.1', '--on_cuda', 'false', '-j', '0',
            '--checkpoint', '/tmp/ft_ci_ckpt_%s' % fed_type,
            '--debug', 'false', '--manual_seed', '7']
    if fed_type == 'fedgate':
        argv += ['--compressed', 'true', '--compressed_ratio', '0.5']
    args = get_args(argv)
    client = Client(args, rank)
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    train_and_validate_federated(client, validate=False)

    # invariant 1: all ranks end with the same model
    flat = client.arena.clone_flat()
    flats = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(flats, flat)
    same = all(torch.equal(flats[0], f) for f in flats)
    q.put((rank, bool(same), float(flat.norm())))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize('fed_type', ['fedavg', 'scaffold', 'fedgate',
                                      'qsparse'])
def test_two_proc_round_consistency(fed_type):
    world = 2
    port = 29700 + abs(hash(fed_type)) % 200
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, fed_type, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    for rank, same, norm in results:
        assert same, 'rank %d diverged' % rank
        assert norm == norm, 'NaN model on rank %d' % rank


def _worker_math(rank, world, port, q):
    """One round of fedavg with known local updates: aggregate must equal
    the mean of client models (scale=1, equal weights)."""
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    dist.init_process_group('gloo', rank=rank, world_size=world)
    import types
    from fedtorch_amd.parallel.arena import Arena
    from fedtorch_amd.parallel.comm import Comm
    from fedtorch_amd.components.optim.sgd import FusedSGD
    from fedtorch_amd.aggregation.federated import fedavg_aggregation
    import torch.nn as nn

    torch.manual_seed(0)
    model = nn.Linear(4, 2)
    arena = Arena(model)
    args = types.SimpleNamespace(
        quantized=False, compressed=False, federated_type='fedavg',
        lr_scale_at_sync=1.0, out_momentum=False, comm_time=[0.0],
        graph=types.SimpleNamespace(rank=rank, n_nodes=world,
                                    on_cuda=False))
    comm = Comm(args)
    server = arena.clone_flat()
    # each rank shifts its model by (rank+1)
    arena.flat.add_(float(rank + 1))
    local = arena.clone_flat()
    opt = FusedSGD(arena, lr=0.1)
    fedavg_aggregation(args, comm, arena, server, opt, list(range(world)))
    # expected: server - mean_diff = server + mean(local - server) = mean(local)
    locals_ = [torch.zeros_like(local) for _ in range(world)]
    dist.all_gather(locals_, local)
    expected = torch.stack(locals_).mean(0)
    ok = torch.allclose(arena.flat, expected, atol=1e-6)
    q.put((rank, bool(ok)))
    dist.barrier()
    dist.destroy_process_group()


def test_fedavg_equals_mean_of_clients():
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_math, args=(r, world, 29950, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert all(ok for _, ok in results)


def _worker_algo(rank, world, port, algo, q):
    try:
        os.environ['MASTER_ADDR'] = '127.0.0.1'
        os.environ['MASTER_PORT'] = str(port)
        os.environ['FEDTORCH_SYNTH_SIZE'] = '200'
        dist.init_process_group('gloo', rank=rank, world_size=world)
        from fedtorch_amd.parameters import get_args
        from fedtorch_amd.main import main as dispatch_main
        argv = ['-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
                '--num_comms', '2', '--online_client_rate', '1.0',
                '-b', '25', '--lr', '0.1', '--on_cuda', 'false',
                '--dist_backend', 'gloo', '-j', '0',
                '--checkpoint', '/tmp/ft_ci_%s' % algo, '--debug', 'false',
                '--manual_seed', '11']
        if algo == 'drfa':
            argv += ['--federated_drfa', 'true', '--federated_type',
                     'fedavg', '--local_step', '3',
                     '--federated_sync_type', 'local_step']
        else:
            argv += ['--federated_type', algo]
        if algo == 'apfl':
            argv += ['--fed_adaptive_alpha', 'true']
        args = get_args(argv)
        dispatch_main(args)
        import torch as _t
        q.put((rank, True))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        import traceback
        traceback.print_exc()
        q.put((rank, False))
        raise


@pytest.mark.parametrize('algo', ['apfl', 'afl', 'drfa', 'packed'])
def test_two_proc_algo_loops(algo):
    world = 2
    port = 29760 + abs(hash(algo)) % 150
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    target = _worker_packed if algo == 'packed' else _worker_algo
    args_t = (world, port, algo, q) if algo != 'packed' else (world, port, q)
    procs = [ctx.Process(target=target, args=(r,) + args_t)
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    assert all(ok for _, ok in results)


def _worker_packed(rank, world, port, q):
    try:
        os.environ['MASTER_ADDR'] = '127.0.0.1'
        os.environ['MASTER_PORT'] = str(port)
        os.environ['FEDTORCH_SYNTH_SIZE'] = '300'
        dist.init_process_group('gloo', rank=rank, world_size=world)
        from fedtorch_amd.parameters import get_args
        from fedtorch_amd.main import main as dispatch_main
        args = get_args([
            '-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
            '--federated_type', 'fedavg', '--num_comms', '2',
            '--online_client_rate', '1.0', '--local_step', '2',
            '--federated_sync_type', 'local_step', '-b', '25',
            '--lr', '0.1', '--on_cuda', 'false', '--dist_backend', 'gloo',
            '-j', '0', '--clients_per_rank', '3', '--in_momentum', 'true',
            '--checkpoint', '/tmp/ft_ci_packed', '--debug', 'false'])
        dispatch_main(args)
        q.put((rank, True))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        import traceback
        traceback.print_exc()
        q.put((rank, False))
        raise


def _worker_i16(rank, world, port, q):
    """int16 quantized payloads must survive
    all_gather_flat (RCCL has no int16 mapping; gathered as byte view)."""
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    dist.init_process_group('gloo', rank=rank, world_size=world)
    import types
    from fedtorch_amd.parallel.comm import Comm
    from fedtorch_amd import ops

    args = types.SimpleNamespace(
        comm_time=[0.0],
        graph=types.SimpleNamespace(rank=rank, n_nodes=world,
                                    on_cuda=False))
    comm = Comm(args)
    torch.manual_seed(42 + rank)
    x = torch.randn(257) * (rank + 1)
    q16, info = comm.all_gather_flat(ops.quantize(x, 16)[0]), None
    # re-quantize locally on every rank and compare against the gathered row
    mine = ops.quantize(x, 16)[0]
    ok = q16.dtype == torch.int16 and q16.shape == (world, 257) and \
        torch.equal(q16[rank], mine)
    q.put((rank, bool(ok)))
    dist.barrier()
    dist.destroy_process_group()


def test_all_gather_int16_payload():
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_i16, args=(r, world, 29971, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert all(ok for _, ok in results)


def _worker_packed_partial(rank, world, port, q):
    """Packed run with BatchNorm + PARTIAL participation (online rate 0.5):
    exercises the weighted BN partials path end to end;
    all ranks must still end bitwise-consistent on params AND BN stats."""
    try:
        os.environ['MASTER_ADDR'] = '127.0.0.1'
        os.environ['MASTER_PORT'] = str(port)
        os.environ['FEDTORCH_SYNTH_SIZE'] = '300'
        dist.init_process_group('gloo', rank=rank, world_size=world)
        from fedtorch_amd.parameters import get_args
        from fedtorch_amd.nodes import Client
        from fedtorch_amd.trainings.packed import (
            train_and_validate_federated_packed)
        from fedtorch_amd.parallel.multiclient import ClientPack
        args = get_args([
            '-d', 'cifar10', '-a', 'resnet20', '-f', 'true',
            '--federated_type', 'fedavg', '--num_comms', '2',
            '--online_client_rate', '0.5', '--local_step', '2',
            '--federated_sync_type', 'local_step', '-b', '25',
            '--lr', '0.1', '--on_cuda', 'false', '--dist_backend', 'gloo',
            '-j', '0', '--clients_per_rank', '3', '--in_momentum', 'true',
            '--checkpoint', '/tmp/ft_ci_packedpp', '--debug', 'false',
            '--manual_seed', '3'])
        client = Client(args, rank)
        client.initialize()
        client.initialize_dataset()
        client.load_local_dataset()
        client.gen_aux_models()
        pack = ClientPack(client, args.clients_per_rank)
        pack.build_loaders()
        train_and_validate_federated_packed(client, pack, validate=False)
        flat = client.arena.clone_flat()
        flats = [torch.zeros_like(flat) for _ in range(world)]
        dist.all_gather(flats, flat)
        same = all(torch.equal(flats[0], f) for f in flats)
        if client.arena.buf_flat is not None:
            buf = client.arena.buf_flat.clone()
            bufs = [torch.zeros_like(buf) for _ in range(world)]
            dist.all_gather(bufs, buf)
            same = same and all(torch.equal(bufs[0], b) for b in bufs)
            # the weighted-partial path must leave FINITE stats
            same = same and bool(torch.isfinite(buf).all())
        else:
            same = False  # resnet MUST expose BN running stats
        q.put((rank, bool(same)))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        import traceback
        traceback.print_exc()
        q.put((rank, False))
        raise


def test_packed_partial_participation_bn():
    world = 2
    ctx = mp.get_context('spawn')
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker_packed_partial,
                         args=(r, world, 29975, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    assert all(ok for _, ok in results)
