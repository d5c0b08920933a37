#!/usr/bin/env python3
# -*- coding: utf-8 -*-
"""Flagship benchmark: ResNet-20 / CIFAR-10 FedAvg local-SGD (BASELINE.json
config 2) on 1..8 MI355X GPUs, one client per GPU rank, tau=10 local steps
between syncs, bf16 compute, synthetic data (no network: BASELINE.md).

A "step" is ONE local SGD step (forward + backward + fused arena SGD); every
tau-th step additionally runs the FedAvg aggregation (weighted arena
all-reduce over RCCL/xGMI) inside the timed region.  Weak scaling: per-GPU
batch is fixed, `value` is the whole-job samples/sec across all ranks.

Launched by the driver as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
import argparse
import json
import os
import sys
import time

# FAST find: same steady-state solver picks as the default hybrid mode on
# this stack (measured 108.2k vs 107.0k samples/s at b256, ab_nhwc.txt) but
# much less exhaustive-probe time on a fresh box — matters when the driver
# launches 8 ranks that all run MIOpen find at once.
os.environ.setdefault('MIOPEN_FIND_MODE', '2')

# Tuned MIOpen user/find-db + compiled-kernel cache shipped in-repo
# (generated on an MI355X box, `scripts/make_miopen_db.sh`): a fresh box
# skips the find phase and any kernel compilation entirely.  MIOpen wants
# write access, so stage a copy under /tmp; per-rank copies avoid 8 ranks
# fighting over sqlite locks.
_REPO = os.path.dirname(os.path.abspath(__file__))
_MDB = os.path.join(_REPO, 'fedtorch_amd', 'miopen_db')
if os.path.isdir(_MDB) and os.environ.get('FEDTORCH_MIOPEN_DB', '1') == '1':
    import shutil
    _r = os.environ.get('RANK', os.environ.get('LOCAL_RANK', '0'))
    _d = '/tmp/ft_miopen_udb_%s_%d' % (_r, os.getpid())
    _c = '/tmp/ft_miopen_cache_%s_%d' % (_r, os.getpid())
    for _sub, _dst in (('udb', _d), ('cache', _c)):
        _src = os.path.join(_MDB, _sub)
        if os.path.isdir(_src):
            shutil.copytree(_src, _dst, dirs_exist_ok=True)
    if os.path.isdir(_d):
        os.environ.setdefault('MIOPEN_USER_DB_PATH', _d)
    if os.path.isdir(_c):
        os.environ.setdefault('MIOPEN_CUSTOM_CACHE_DIR', _c)

import torch
import torch.distributed as dist

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

from fedtorch_amd.parameters import get_args  # noqa: E402
from fedtorch_amd.nodes import Client  # noqa: E402
from fedtorch_amd.trainings.federated import amp  # noqa: E402
from fedtorch_amd.aggregation.federated import (  # noqa: E402
    fedavg_aggregation, fedgate_aggregation, aggregate_bn_buffers)

TAU = 10  # local steps per communication round (BASELINE config 2)


def parse():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=100)
    p.add_argument('--warmup', type=int, default=20)
    p.add_argument('--batch', type=int, default=256)
    p.add_argument('--model', type=str, default='resnet20')
    p.add_argument('--dtype', type=str, default='bf16')
    p.add_argument('--graph', type=str, default='auto',
                   help='hipGraph-capture the local step: auto|on|off')
    # NHWC end-to-end (custom NHWC fused-BN + stem kernels dodge MIOpen's
    # naive bf16 fallbacks) measured +23% over NCHW at b256; see
    # profiles/r01_bench_notes.md
    p.add_argument('--layout', type=str, default='nhwc',
                   choices=['nhwc', 'nchw'])
    p.add_argument('--bf16_weights', type=str, default='on',
                   choices=['on', 'off'],
                   help='bf16 compute twin for conv/linear weights (fp32 '
                        'master in the arena; graph mode only)')
    p.add_argument('--fused_bn', type=str, default='on',
                   choices=['on', 'off'])
    p.add_argument('--fp32_stem', type=str, default='off',
                   choices=['on', 'off'])
    p.add_argument('--clients', type=int, default=1,
                   help='virtual clients per rank (packed mode when > 1: '
                        'replicas resident in HBM, BASELINE config 5)')
    p.add_argument('--streams', type=int, default=1,
                   help='concurrent HIP streams (packed mode). Measured on '
                        'MI355X: multi-stream replay does NOT overlap for '
                        'this kernel mix (profiles/r01_bench_notes.md), so '
                        'the default is 1.')
    p.add_argument('--algo', type=str, default='fedavg',
                   choices=['fedavg', 'comgate_topk', 'comgate_quant'],
                   help='sync algorithm: plain FedAvg or FedCOMGATE '
                        '(FedGATE + top-k compression / quantization, '
                        'BASELINE config 3)')
    return p.parse_args()


def main():
    b = parse()
    on_gpu = torch.cuda.is_available()
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', str(b.gpus)))
    if world > 1 or 'RANK' in os.environ:
        os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
        os.environ.setdefault('MASTER_PORT', '29617')
        dist.init_process_group('cpu:gloo,cuda:nccl' if on_gpu else 'gloo')
        rank = dist.get_rank()
        world = dist.get_world_size()

    use_bf16 = b.dtype == 'bf16' and on_gpu
    fed_type = 'fedgate' if b.algo.startswith('comgate') else 'fedavg'
    argv_extra = []
    if b.algo == 'comgate_topk':
        argv_extra = ['--compressed', 'true', '--compressed_ratio', '0.2']
    elif b.algo == 'comgate_quant':
        argv_extra = ['--quantized', 'true', '--quantized_bits', '8']
    args = get_args(argv_extra + [
        '-d', 'cifar10', '-a', b.model, '-f', 'true',
        '--federated_type', fed_type, '--num_comms', '1000000',
        '--online_client_rate', '1.0', '--federated_sync_type', 'local_step',
        '--local_step', str(TAU), '-b', str(b.batch), '--lr', '0.1',
        '--in_momentum', 'true', '--weight_decay', '5e-4',
        '--bf16', 'true' if use_bf16 else 'false',
        '--channels_last',
        'true' if (on_gpu and b.layout == 'nhwc') else 'false',
        '--fused_bn', 'true' if b.fused_bn == 'on' else 'false',
        '-j', '0', '--checkpoint', '/tmp/ft_bench_ckpt', '--debug', 'false'])
    os.environ.setdefault('FEDTORCH_SYNTH_SIZE', '2048')

    if on_gpu:
        import fedtorch_amd.ops as ops
        if not ops.hip_available():
            raise RuntimeError('HIP kernel pack not built: bench refuses the '
                               'eager fallback on GPU')

    client = Client(args, rank)
    client.initialize()
    client.gen_aux_models()
    if b.fp32_stem == 'on' and on_gpu:
        from fedtorch_amd.components.models.resnet import _Fp32Stem
        client.model.conv1 = _Fp32Stem(client.model.conv1)
    args = client.args
    # pretend dataset-derived counters (synthetic pool below replaces loaders)
    args.num_batches_train_per_device_per_epoch = 390
    device = torch.device('cuda') if on_gpu else torch.device('cpu')

    # synthetic CIFAR-shaped data pool, resident on device
    g = torch.Generator(device='cpu').manual_seed(1234 + rank)
    pool_n = 8
    xs = torch.randn((pool_n, b.batch, 3, 32, 32), generator=g)
    ys = torch.randint(0, 10, (pool_n, b.batch), generator=g)
    ys = ys.to(device)
    if args.channels_last and on_gpu:
        xs = [xs[i].to(device).to(memory_format=torch.channels_last)
              for i in range(pool_n)]
    else:
        xs = xs.to(device)

    lr = 0.1
    for pg in client.optimizer.param_groups:
        pg['lr'] = lr
    online = list(range(world))
    client.model.train()

    if b.clients > 1:
        return run_packed_bench(b, client, args, xs, ys, pool_n, world,
                                rank, on_gpu)

    def inner(x, y):
        """one local SGD step: fwd + loss + bwd + fused arena step.
        NO metrics / .item() in the hot loop (those are logging, reference
        computes them per step only for console output)."""
        client.optimizer.zero_grad()
        with amp(args):
            loss = client.criterion(client.model(x), y)
        loss.backward()
        client.optimizer.step(apply_lr=True, apply_in_momentum=True,
                              apply_out_momentum=False)

    use_graph = on_gpu and b.graph != 'off'
    if use_graph:
        try:
            arena = client.arena
            if b.bf16_weights == 'on' and use_bf16:
                # conv/linear params live in a bf16 twin arena: no autocast
                # weight-cast kernels per step; the fused SGD refreshes the
                # twin and the grad gather casts bf16 grads back to fp32.
                arena.enable_bf16_compute()

            def inner_stolen(x, y):
                arena.detach_grads()
                with amp(args):
                    loss = client.criterion(client.model(x), y)
                loss.backward()
                arena.gather_grads()
                client.optimizer.step(apply_lr=True, apply_in_momentum=True,
                                      apply_out_momentum=False)

            static_x = (xs[0] if isinstance(xs, list) else xs[0]).clone()
            static_y = ys[0].clone()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    inner_stolen(static_x, static_y)
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            # Two captures with stolen grads: with p.grad=None at capture,
            # AccumulateGrad takes ownership of the produced buffer instead
            # of launching one fp32 add per parameter per step (~65 kernels
            # on ResNet-20); graph2 gathers the stolen buffers into the
            # contiguous grad arena with ONE kernel, then steps.
            arena.detach_grads()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                with amp(args):
                    loss = client.criterion(client.model(static_x), static_y)
                loss.backward()
            arena.gather_grads()  # builds the chunk table outside capture
            graph2 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph2):
                arena.gather_grads()
                client.optimizer.step(apply_lr=True, apply_in_momentum=True,
                                      apply_out_momentum=False)
        except Exception as e:  # noqa: BLE001
            if b.graph == 'on':
                raise
            print('[bench] hipGraph capture failed (%r), eager path' % e,
                  flush=True)
            use_graph = False
            client.arena.disable_bf16_compute()
            client.arena.attach_grads()
    if use_graph:
        def local_step(i):
            static_x.copy_(xs[i % pool_n])
            static_y.copy_(ys[i % pool_n])
            graph.replay()
            graph2.replay()
    else:
        def local_step(i):
            inner(xs[i % pool_n], ys[i % pool_n])

    def sync():
        args.comm_time.append(0.0)
        if b.algo.startswith('comgate'):
            fedgate_aggregation(args, client.comm, client.arena,
                                client.model_server, client.model_delta,
                                client.model_memory, client.optimizer,
                                online, lr, TAU, work=client.work)
        else:
            fedavg_aggregation(args, client.comm, client.arena,
                               client.model_server, client.optimizer,
                               online, work=client.work)
        aggregate_bn_buffers(args, client.comm, client.arena, online,
                             work=client.work)
        client.arena.sync_half()

    def run(n_steps):
        for s in range(1, n_steps + 1):
            local_step(s)
            if s % TAU == 0:
                sync()

    # ---- settle (untimed setup, independent of --warmup) ----
    # A fresh box pays one-time costs on the FIRST execution of each
    # kernel/collective (HIP code-object load, allocator growth, RCCL
    # channel setup, MIOpen find leftovers).  With the driver's
    # --warmup 5 < tau no sync would run before timing and the whole
    # first-round cost lands inside a ~60 ms timed window (round-1
    # driver bench measured 3.1 ms/step vs 1.64 steady for exactly this
    # reason).  So: run full rounds here until per-round time stabilizes
    # (<=8 rounds), then do the W contractual warmup steps.
    if on_gpu or os.environ.get('FEDTORCH_FORCE_SETTLE') == '1':
        prev = None
        for r_i in range(8):
            if on_gpu:
                torch.cuda.synchronize()
            ts = time.perf_counter()
            run(TAU)
            if on_gpu:
                torch.cuda.synchronize()
            dt = time.perf_counter() - ts
            if dist.is_initialized():
                # all ranks must take the SAME number of settle rounds
                # (sync() is collective): agree on the MAX round time
                t_ = torch.tensor([dt], dtype=torch.float64)
                dist.all_reduce(t_, op=dist.ReduceOp.MAX)
                dt = float(t_[0])
            if prev is not None and r_i >= 1 and dt < prev * 1.10:
                break
            prev = dt

    # ---- warmup ----
    run(b.warmup)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()

    # ---- timed ----
    t0 = time.perf_counter()
    run(b.steps)
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = float(t[0])

    if rank == 0:
        value = world * b.batch * b.steps / elapsed
        out = {
            'metric': 'samples/sec',
            'value': value,
            'unit': 'samples/sec',
            'n_gpus': world,
            'steps': b.steps,
            'warmup': b.warmup,
            'ms_per_step': elapsed / b.steps * 1e3,
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': 'bf16' if use_bf16 else 'fp32',
            'data': 'synthetic',
            'config': {'model': b.model, 'global_batch': world * b.batch,
                       'seq_len': None,
                       'parallelism': '%s_dp%d_tau%d' % (b.algo, world,
                                                         TAU)},
        }
        print(json.dumps(out), flush=True)
    if dist.is_initialized():
        dist.destroy_process_group()



def run_packed_bench(b, client, args, xs, ys, pool_n, world, rank, on_gpu):
    """Packed virtual clients: C client states per rank, K = --streams
    model replicas executing CONCURRENTLY on separate HIP streams (each
    slot replays its own captured graph).  Aggregation = one batched
    multi_diff_accumulate over the [C, N] replica arena + one all-reduce.
    """
    from copy import deepcopy
    from fedtorch_amd.parallel.arena import Arena
    from fedtorch_amd.components.optim.sgd import FusedSGD
    from fedtorch_amd import ops as ft_ops

    C = b.clients
    K = min(b.streams, C) if on_gpu else 1
    device = client.arena.flat.device
    n = client.arena.numel
    g0 = client.optimizer.param_groups[0]

    # per-client resident state
    state = torch.zeros((C, n), device=device)
    mom = torch.zeros((C, n), device=device)
    for c in range(C):
        state[c].copy_(client.arena.flat)
    server = client.arena.clone_flat()
    weights = torch.full((C,), 1.0 / (C * world), device=device)
    partial = torch.zeros(n, device=device)

    # K execution slots: independent module replicas + arenas + optimizers
    slots = []
    for k in range(K):
        m = deepcopy(client.model)
        arena = Arena(m)
        opt = FusedSGD(arena, lr=g0['lr'], in_momentum=g0['in_momentum'],
                       weight_decay=g0['weight_decay'])
        sx = (xs[0] if isinstance(xs, list) else xs[0]).clone()
        sy = ys[0].clone()
        stream = torch.cuda.Stream() if on_gpu else None
        slots.append(dict(model=m, arena=arena, opt=opt, sx=sx, sy=sy,
                          stream=stream, graph=None))

    def slot_step(sl):
        sl['opt'].zero_grad()
        with amp(args):
            loss = client.criterion(sl['model'](sl['sx']), sl['sy'])
        loss.backward()
        sl['opt'].step(apply_lr=True, apply_in_momentum=True,
                       apply_out_momentum=False)

    def slot_step_stolen(sl):
        # stolen-grad flow (same scheme as the single-client graph path)
        sl['arena'].detach_grads()
        with amp(args):
            loss = client.criterion(sl['model'](sl['sx']), sl['sy'])
        loss.backward()
        sl['arena'].gather_grads()
        sl['opt'].step(apply_lr=True, apply_in_momentum=True,
                       apply_out_momentum=False)

    if on_gpu and b.graph != 'off':
        if b.bf16_weights == 'on' and args.bf16:
            for sl in slots:
                sl['arena'].enable_bf16_compute()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for sl in slots:
                for _ in range(3):
                    slot_step_stolen(sl)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        for sl in slots:
            sl['arena'].detach_grads()
            g1 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g1):
                with amp(args):
                    loss = client.criterion(sl['model'](sl['sx']), sl['sy'])
                loss.backward()
            sl['arena'].gather_grads()
            g2 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g2):
                sl['arena'].gather_grads()
                sl['opt'].step(apply_lr=True, apply_in_momentum=True,
                               apply_out_momentum=False)
            sl['graph'] = (g1, g2)

    main_stream = torch.cuda.current_stream() if on_gpu else None

    # hipGraph + momentum caveat: a captured graph bakes in the slot's
    # momentum buffer POINTER, so per-client momentum rows are STAGED into
    # the slot buffer around each client's tau steps (copies are ~N floats,
    # trivial next to the steps themselves).
    for sl in slots:
        if sl['opt']._in_buf is None:
            sl['opt'].bind_state(in_buf=torch.zeros(n, device=device),
                                 in_init=True)

    def run_round_staged(r):
        for base in range(0, C, K):
            group = list(range(base, min(base + K, C)))
            if on_gpu:
                for k, c in enumerate(group):
                    sl = slots[k]
                    sl['stream'].wait_stream(main_stream)
                    with torch.cuda.stream(sl['stream']):
                        sl['arena'].flat.copy_(server)
                        sl['arena'].sync_half()
                        sl['opt']._in_buf.copy_(mom[c])
                        for t in range(TAU):
                            sl['sx'].copy_(xs[(c + t) % pool_n])
                            sl['sy'].copy_(ys[(c + t) % pool_n])
                            if sl['graph'] is not None:
                                sl['graph'][0].replay()
                                sl['graph'][1].replay()
                            else:
                                slot_step(sl)
                        state[c].copy_(sl['arena'].flat)
                        mom[c].copy_(sl['opt']._in_buf)
                    main_stream.wait_stream(sl['stream'])
            else:
                for k, c in enumerate(group):
                    sl = slots[k]
                    sl['arena'].flat.copy_(server)
                    sl['opt']._in_buf.copy_(mom[c])
                    for t in range(TAU):
                        sl['sx'].copy_(xs[(c + t) % pool_n])
                        sl['sy'].copy_(ys[(c + t) % pool_n])
                        slot_step(sl)
                    state[c].copy_(sl['arena'].flat)
                    mom[c].copy_(sl['opt']._in_buf)
        ft_ops.multi_diff_accumulate(server, state, weights, partial)
        client.comm.all_reduce(partial)
        server.sub_(partial)

    rounds_warm = max(b.warmup // TAU, 1)
    rounds_timed = max(b.steps // TAU, 1)
    for r in range(rounds_warm):
        run_round_staged(r)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for r in range(rounds_timed):
        run_round_staged(r + rounds_warm)
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64)
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t[0])
    if rank == 0:
        total_steps = rounds_timed * TAU * C
        value = world * b.batch * total_steps / elapsed
        out = {
            'metric': 'samples/sec', 'value': value, 'unit': 'samples/sec',
            'n_gpus': world, 'steps': rounds_timed * TAU,
            'warmup': rounds_warm * TAU,
            'ms_per_step': elapsed / (rounds_timed * TAU) * 1e3,
            'higher_is_better': True, 'scaling': 'weak',
            'vs_baseline': None,
            'dtype': 'bf16' if args.bf16 else 'fp32', 'data': 'synthetic',
            'config': {'model': b.model,
                       'global_batch': world * b.batch * C,
                       'seq_len': None,
                       'parallelism': 'fedavg_packed_c%d_k%d_dp%d_tau%d' % (
                           C, K, world, TAU)},
        }
        print(json.dumps(out), flush=True)
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
