# -*- coding: utf-8 -*-
"""Distributed entry (parity with reference `main.py`).

Launch with torchrun, one process per MI355X GPU:
    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 -m fedtorch_amd.main -f true ...
Backend: nccl (= RCCL over xGMI) on GPU, gloo on CPU.
"""
import os

import torch
import torch.distributed as dist

from fedtorch_amd.parameters import get_args
from fedtorch_amd.nodes import Client
from fedtorch_amd.trainings.local_sgd import train_and_validate
from fedtorch_amd.trainings.federated import train_and_validate_federated


def init_distributed(args):
    """Init the process group; returns True if THIS call created it (the
    caller then owns teardown — an externally created group is left
    alone)."""
    if dist.is_initialized():
        return False
    backend = args.dist_backend
    if backend in ('mpi', None):
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    if backend == 'nccl' and not torch.cuda.is_available():
        backend = 'gloo'
    if backend == 'nccl':
        # RCCL for GPU tensors, gloo for the small host-side control
        # messages (online-client ids, dataset indices, metric averages).
        backend = 'cpu:gloo,cuda:nccl'
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29500')
    os.environ.setdefault('RANK', '0')
    os.environ.setdefault('WORLD_SIZE', '1')
    dist.init_process_group(backend)
    return True


def main(args):
    owns_pg = init_distributed(args)
    try:
        _dispatch(args)
    finally:
        if owns_pg and dist.is_initialized():
            dist.destroy_process_group()


def _dispatch(args):
    client = Client(args, dist.get_rank())
    client.initialize()
    client.initialize_dataset()
    client.load_local_dataset()
    client.gen_aux_models()
    if args.federated and args.clients_per_rank > 1:
        # virtual-client packing: C clients per GPU rank, replicas resident
        # in HBM (fedavg / DRFA-over-fedavg).
        from fedtorch_amd.parallel.multiclient import ClientPack
        from fedtorch_amd.trainings.packed import \
            train_and_validate_federated_packed
        pack = ClientPack(client, args.clients_per_rank)
        pack.build_loaders()
        train_and_validate_federated_packed(client, pack)
        return
    if args.federated:
        if args.federated_drfa:
            from fedtorch_amd.trainings.drfa import \
                train_and_validate_federated_drfa
            train_and_validate_federated_drfa(client)
        elif args.federated_type == 'apfl':
            from fedtorch_amd.trainings.apfl import \
                train_and_validate_federated_apfl
            train_and_validate_federated_apfl(client)
        elif args.federated_type == 'afl':
            from fedtorch_amd.trainings.afl import \
                train_and_validate_federated_afl
            train_and_validate_federated_afl(client)
        elif args.federated_type in ('fedavg', 'scaffold', 'fedgate',
                                     'qsparse', 'fedprox', 'fedadam'):
            train_and_validate_federated(client)
        else:
            raise NotImplementedError(args.federated_type)
    else:
        train_and_validate(client)


if __name__ == '__main__':
    main(get_args())
