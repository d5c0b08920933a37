# -*- coding: utf-8 -*-
"""Run-time initialization (parity with reference `utils/init_config.py`).

Builds the logical graph, binds rank -> GPU (one process per MI355X via
LOCAL_RANK), seeds, zeroes the run-time counters the training loops maintain
(`init_config.py:10-20`), initializes checkpoint dirs and the sync scheme.
"""
import random

import numpy as np
import torch

from fedtorch_amd.utils.topology import FCGraph
from fedtorch_amd.logs.checkpoint import init_checkpoint
from fedtorch_amd.logs.logging import configure_log


def init_runtime_state(args):
    """Zero the run-time counters (reference `init_config.py:10-20`)."""
    args.local_index = 0
    args.client_epoch_total = 0
    args.block_index = 0
    args.global_index = 0
    args.local_data_seen = 0
    args.best_prec1 = 0
    args.best_epoch = []
    args.rounds_comm = 0
    args.comm_time = []
    args.epoch_ = 0.0
    args.epoch = 0


def init_config(args, rank=None):
    if rank is None:
        import torch.distributed as dist
        rank = dist.get_rank() if (dist.is_available() and
                                   dist.is_initialized()) else 0
    args.graph = FCGraph(rank, args.blocks, args.on_cuda and
                         torch.cuda.is_available(), args.world)
    # only the server (rank 0) logs by default (reference
    # `init_config.py:28-30`).
    if rank != 0:
        args.debug = False

    init_runtime_state(args)

    if args.graph.on_cuda:
        torch.cuda.set_device(args.graph.device)
        torch.backends.cudnn.benchmark = True  # MIOpen auto-tune
    torch.manual_seed(args.manual_seed)
    np.random.seed(args.manual_seed)
    random.seed(args.manual_seed)
    if args.graph.on_cuda:
        torch.cuda.manual_seed(args.manual_seed)

    init_checkpoint(args)
    configure_log(args)
    from fedtorch_amd.aggregation.distributed import configure_sync_scheme
    configure_sync_scheme(args)
    return args


def init_config_centered(args, rank=0):
    """Single-process simulation: everything on cuda:0 (reference
    `init_config.py:64`)."""
    # In centered (single-process simulation) mode the reference reuses
    # --num_workers as the number of simulated clients (`main_centered.py:20`).
    args.graph = FCGraph(rank, args.blocks,
                         args.on_cuda and torch.cuda.is_available(),
                         args.world, n_nodes=args.num_workers)
    args.debug = args.debug or rank == 0
    init_runtime_state(args)
    if args.graph.on_cuda:
        torch.cuda.set_device(0)
        torch.backends.cudnn.benchmark = True
    torch.manual_seed(args.manual_seed)
    np.random.seed(args.manual_seed)
    random.seed(args.manual_seed)
    init_checkpoint(args)
    configure_log(args)
    from fedtorch_amd.aggregation.distributed import configure_sync_scheme
    configure_sync_scheme(args)
    return args
