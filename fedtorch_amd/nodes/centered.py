# -*- coding: utf-8 -*-
"""Centered (single-process) nodes — parity with reference
`fedtorch/nodes/nodes_centered.py:27-213`.

This is the mode the MI355X build generalizes to virtual-client packing:
every client's replica + optimizer state + aux buffers stay resident in the
GPU's 288 GB HBM3E, and the sequential client loop runs fused arena kernels
per client (batched multi-client kernels are the next step — see
`fedtorch_amd/parallel/multiclient.py`).
"""
from copy import copy, deepcopy

import torch

from fedtorch_amd.components.comps import create_components
from fedtorch_amd.components.optimizer import define_optimizer
from fedtorch_amd.components.dataset import define_dataset
from fedtorch_amd.parallel.arena import Arena
from fedtorch_amd.utils.init_config import init_config_centered
from fedtorch_amd.logs.logging import configure_log, log_args
from fedtorch_amd.logs.meter import define_val_tracker
from fedtorch_amd.aggregation.distributed import configure_sync_scheme
from fedtorch_amd.nodes.client import Node


class ClientCentered(Node):
    def __init__(self, args, rank, Partitioner=None):
        super().__init__(rank)
        self.args = copy(args)
        self.Partitioner = None
        self.work = {}
        self.initialize()
        self.load_local_dataset(Partitioner)
        self.gen_aux_models()
        self.local_val_tracker = define_val_tracker()
        self.global_val_tracker = define_val_tracker()
        if self.args.fed_personal:
            self.local_personal_val_tracker = define_val_tracker()
            self.global_personal_val_tracker = define_val_tracker()
        if self.args.federated_sync_type == 'epoch':
            self.args.local_step = \
                self.args.num_epochs_per_comm * len(self.train_loader)
            configure_sync_scheme(self.args)

    def initialize(self):
        init_config_centered(self.args, self.rank)
        (self.model, self.arena, self.criterion, self.scheduler,
         self.optimizer, self.metrics) = create_components(self.args)
        self.args.finish_one_epoch = False
        if self.rank == 0:
            configure_log(self.args)
            log_args(self.args, debug=self.args.debug)

    def make_model_consistent(self, ref_arena):
        self.arena.load_flat(ref_arena.flat)

    def load_local_dataset(self, Partitioner):
        args = self.args
        per_client_data = args.data in ('emnist', 'emnist_full', 'synthetic',
                                        'shakespeare')
        want_partitioner = (not per_client_data) and self.rank == 0
        kw = dict(shuffle=True, test=False)
        if per_client_data:
            loaders = define_dataset(args, **kw)
        elif want_partitioner:
            loaders, self.Partitioner = define_dataset(
                args, return_partitioner=True, **kw)
        else:
            loaders = define_dataset(args, Partitioner=Partitioner, **kw)
        if args.fed_personal:
            if args.federated_type == 'perfedavg':
                (self.train_loader, self.test_loader, self.val_loader,
                 self.val_loader1) = loaders
            else:
                self.train_loader, self.test_loader, self.val_loader = loaders
        else:
            self.train_loader, self.test_loader = loaders
        if args.data in ('mnist', 'fashion_mnist', 'cifar10'):
            args.classes = torch.arange(10)
        elif args.data == 'synthetic':
            args.classes = torch.arange(5)
        elif args.data == 'adult':
            args.classes = torch.arange(2)

    def gen_aux_models(self):
        args = self.args
        new = self.arena.new_buffer
        if not args.federated:
            return
        t = args.federated_type
        if t == 'fedgate':
            self.model_delta = new()
            self.model_memory = new()
        elif t == 'qsparse':
            self.model_memory = new()
        elif t == 'scaffold':
            self.model_client_control = new()
        elif t == 'fedadam':
            args.fedadam_v = [args.fedadam_tau ** 2] * len(self.arena.params)
        elif t in ('apfl', 'perfedme'):
            self.model_personal = deepcopy(self.model)
            self.arena_personal = Arena(self.model_personal)
            self.optimizer_personal = define_optimizer(args,
                                                       self.arena_personal)
        elif t == 'qffl':
            self.full_loss = 0.0
        if args.federated_drfa:
            self.kth_model = new()

    def zero_avg(self):
        self.model_avg = self.arena.new_buffer()
        self.model_avg_tmp = self.arena.new_buffer()


class ServerCentered(Node):
    def __init__(self, args, model_server, rank=0):
        super().__init__(0)
        self.args = copy(args)
        self.args.epoch = 1
        self.rnn = self.args.arch == 'rnn'
        self.work = {}
        self.initialize()
        self.gen_aux_models()
        self.local_val_tracker = define_val_tracker()
        self.global_val_tracker = define_val_tracker()
        if self.args.fed_personal:
            self.local_personal_val_tracker = define_val_tracker()
            self.global_personal_val_tracker = define_val_tracker()
        self.global_test_tracker = define_val_tracker()
        self.load_test_dataset()
        self.grad = self.arena.new_buffer()

    def initialize(self):
        (self.model, self.arena, self.criterion, self.scheduler,
         self.optimizer, self.metrics) = create_components(self.args)

    def zero_grad(self):
        self.grad.zero_()

    def zero_avg(self):
        self.model_avg = self.arena.new_buffer()

    def update_model(self):
        """p -= lr_scale_at_sync * accumulated grad (reference
        `nodes_centered.py:176-179`)."""
        self.arena.flat.add_(self.grad, alpha=-self.args.lr_scale_at_sync)

    def enable_grad(self, dataloader):
        """Arena grads are pre-materialized, kept for API parity (reference
        `nodes_centered.py:181-191`)."""
        self.optimizer.zero_grad()

    def gen_aux_models(self):
        args = self.args
        if not args.federated:
            return
        if args.federated_type == 'scaffold':
            self.model_server_control = self.arena.new_buffer()
        elif args.federated_type == 'afl':
            self.lambda_vector = torch.zeros(args.graph.n_nodes)
        if args.federated_drfa:
            self.kth_model = self.arena.new_buffer()
            self.lambda_vector = torch.zeros(args.graph.n_nodes)

    def load_test_dataset(self):
        loaders = define_dataset(self.args, shuffle=True, test=True)
        self.test_loader = loaders[1]
