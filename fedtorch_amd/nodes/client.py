# -*- coding: utf-8 -*-
"""Client node (parity with reference `fedtorch/nodes/nodes.py:29-116`;
public API kept: initialize / initialize_dataset / load_local_dataset /
gen_aux_models, per reference README.md:59-71).

MI355X-native differences:
* parameters live in a flat :class:`Arena`; ``model_server`` and every
  auxiliary model (FedGATE delta/memory, SCAFFOLD control variates, DRFA kth
  model) are flat arena-shaped buffers resident in HBM3E — the reference
  deep-copies entire nn.Modules for each (`nodes.py:87-112`);
* no per-round ``dist.new_group`` (`nodes.py:62`): the world communicator is
  cached and weighted collectives handle partial participation.
"""
import platform
from copy import deepcopy, copy

import torch

from fedtorch_amd.components.comps import create_components
from fedtorch_amd.components.optimizer import define_optimizer
from fedtorch_amd.components.model import consistent_arena
from fedtorch_amd.components.dataset import define_dataset
from fedtorch_amd.components.datasets.prepare_data import get_dataset
from fedtorch_amd.parallel.arena import Arena
from fedtorch_amd.parallel.comm import Comm
from fedtorch_amd.utils.init_config import init_config
from fedtorch_amd.logs.logging import log, configure_log, log_args


class Node(object):
    def __init__(self, rank):
        self.rank = rank

    def initialize(self):
        pass

    def reset_tracker(self, tracker):
        for k in tracker.keys():
            tracker[k].reset()


class Client(Node):
    def __init__(self, args, rank):
        super().__init__(rank)
        self.args = copy(args)

    def initialize(self):
        init_config(self.args, self.rank)
        (self.model, self.arena, self.criterion, self.scheduler,
         self.optimizer, self.metrics) = create_components(self.args)
        self.args.finish_one_epoch = False
        self.comm = Comm(self.args)
        self.work = {}
        # one flat broadcast makes rank 0's init authoritative (reference
        # does P all-reduces, `components/model.py:33-43`).
        consistent_arena(self.arena)
        # flat copy of the server model (reference deep-copies the module,
        # `nodes.py:48`).
        self.model_server = self.arena.clone_flat()
        configure_log(self.args)
        log_args(self.args, debug=self.args.debug)
        log('Rank {} with block {} on {} {}-{}'.format(
            self.args.graph.rank,
            self.args.graph.ranks_with_blocks[self.args.graph.rank],
            platform.node(),
            'GPU' if self.args.graph.on_cuda else 'CPU',
            self.args.graph.device), debug=self.args.debug)

    def initialize_dataset(self):
        """rank 0 materializes the dataset first; everyone then proceeds
        (reference `nodes.py:64-71`; nothing downloads in this environment,
        generation/parsing happens on first touch)."""
        if self.args.graph.rank == 0:
            get_dataset(self.args, self.args.data, self.args.data_dir,
                        split='train')
            get_dataset(self.args, self.args.data, self.args.data_dir,
                        split='test')
        self.comm.barrier()

    def load_local_dataset(self):
        load_test = self.args.graph.rank == 0
        loaders = define_dataset(self.args, shuffle=True, test=load_test)
        if self.args.fed_personal:
            if self.args.federated_type == 'perfedavg':
                (self.train_loader, self.test_loader, self.val_loader,
                 self.val_loader1) = loaders
            else:
                self.train_loader, self.test_loader, self.val_loader = loaders
        else:
            self.train_loader, self.test_loader = loaders
        if self.args.data in ('mnist', 'fashion_mnist', 'cifar10', 'cifar100'):
            self.args.classes = torch.arange(10)
        elif self.args.data == 'synthetic':
            self.args.classes = torch.arange(5)
        elif self.args.data == 'adult':
            self.args.classes = torch.arange(2)

    def gen_aux_models(self):
        """Per-algorithm auxiliary state as flat buffers (reference
        `nodes.py:87-112`)."""
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
            self.model_server_control = new()
        elif t == 'fedadam':
            args.fedadam_v = [args.fedadam_tau ** 2] * len(self.arena.params)
        elif t in ('apfl', 'perfedme'):
            self.model_personal = deepcopy(self.model)
            self.arena_personal = Arena(self.model_personal)
            self.optimizer_personal = define_optimizer(args,
                                                       self.arena_personal)
        elif t == 'afl':
            self.lambda_vector = torch.zeros(args.graph.n_nodes)
        elif t == 'qffl':
            self.full_loss = 0.0
        if args.federated_drfa:
            self.kth_model = new()
            self.lambda_vector = torch.zeros(args.graph.n_nodes)

    def zero_avg(self):
        self.model_avg = self.arena.new_buffer()
        self.model_avg_tmp = self.arena.new_buffer()
