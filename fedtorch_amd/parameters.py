# -*- coding: utf-8 -*-
"""Flag surface for fedtorch_amd.

Keeps the reference CLI contract (reference `fedtorch/parameters.py:12-260`:
same flag names, defaults and post-parse derivations) while adding a small
set of MI355X-specific flags (``--bf16``, ``--hip_kernels``, ``--hip_graph``,
``--clients_per_rank``).  Unlike the reference, run-time counters do NOT live
on the argparse namespace; they live on a ``TrainState`` (see
`fedtorch_amd/utils/init_config.py`) — the namespace stays configuration-only,
but the same attribute names are mirrored for tooling parity.
"""
import argparse
import time


MODEL_NAMES = sorted([
    'logistic_regression', 'least_square', 'robust_logistic_regression',
    'robust_least_square', 'mlp', 'robust_mlp', 'cnn', 'rnn',
    'resnet8', 'resnet14', 'resnet20', 'resnet32', 'resnet44', 'resnet56',
    'resnet110', 'resnet18', 'resnet34', 'resnet50', 'resnet101', 'resnet152',
    'densenet', 'wideresnet',
])

DATASET_NAMES = ['cifar10', 'cifar100', 'mnist', 'fashion_mnist',
                 'emnist', 'emnist_full', 'synthetic', 'shakespeare', 'adult',
                 'epsilon', 'MSD', 'higgs', 'rcv1', 'stl10']

FEDERATED_TYPES = ['fedavg', 'scaffold', 'fedprox', 'fedgate', 'fedadam',
                   'apfl', 'afl', 'perfedavg', 'qsparse', 'perfedme', 'qffl']


def str2bool(v):
    """Convert bool-ish strings to bool (reference `parameters.py:263-280`)."""
    if isinstance(v, bool):
        return v
    if v.lower() in ('yes', 'true', 't', 'y', '1'):
        return True
    if v.lower() in ('no', 'false', 'f', 'n', '0'):
        return False
    raise argparse.ArgumentTypeError('Boolean value expected.')


def build_parser():
    parser = argparse.ArgumentParser(
        description='Parameters for training with the fedtorch_amd engine.')

    # dataset.
    parser.add_argument('-d', '--data', default='cifar10', choices=DATASET_NAMES,
                        help='Dataset name.')
    parser.add_argument('-p', '--data_dir', default='./data/',
                        help='path to dataset')
    parser.add_argument('--partition_data', default=True, type=str2bool,
                        help='decide if each worker will access to all data.')
    parser.add_argument('--pin_memory', default=True, type=str2bool)
    parser.add_argument('--synthetic_alpha', default=0.0, type=float)
    parser.add_argument('--synthetic_beta', default=0.0, type=float)
    parser.add_argument('--sensitive_feature', default=9, type=int)

    # federated setting.
    parser.add_argument('-f', '--federated', default=False, type=str2bool)
    parser.add_argument('--num_class_per_client', default=1, type=int)
    parser.add_argument('--num_comms', default=100, type=int)
    parser.add_argument('--online_client_rate', default=0.1, type=float)
    parser.add_argument('--federated_sync_type', default='epoch', type=str,
                        choices=['epoch', 'local_step'])
    parser.add_argument('--num_epochs_per_comm', default=1, type=int)
    parser.add_argument('--iid_data', default=True, type=str2bool)
    parser.add_argument('--federated_type', default='fedavg', type=str,
                        choices=FEDERATED_TYPES)
    parser.add_argument('--unbalanced', default=False, type=str2bool)
    parser.add_argument('--dirichlet', default=False, type=str2bool)
    parser.add_argument('--fed_personal', default=False, type=str2bool)
    parser.add_argument('--fed_personal_alpha', default=0.5, type=float)
    parser.add_argument('--fed_adaptive_alpha', default=False, type=str2bool)
    parser.add_argument('--fed_personal_test', default=False, type=str2bool)
    parser.add_argument('--fedadam_beta', default=0.9, type=float)
    parser.add_argument('--fedadam_tau', default=0.1, type=float)
    parser.add_argument('--quantized', default=False, type=str2bool)
    parser.add_argument('--quantized_bits', default=8, type=int)
    parser.add_argument('--compressed', default=False, type=str2bool)
    parser.add_argument('--compressed_ratio', default=1.0, type=float)
    parser.add_argument('--federated_drfa', default=False, type=str2bool)
    parser.add_argument('--drfa_gamma', default=0.1, type=float)
    parser.add_argument('--per_class_acc', default=False, type=str2bool)
    parser.add_argument('--perfedavg_beta', default=0.001, type=float)
    parser.add_argument('--fedprox_mu', default=0.002, type=float)
    parser.add_argument('--perfedme_lambda', default=15, type=float)
    parser.add_argument('--qffl_q', default=0.0, type=float)

    # model.
    parser.add_argument('-a', '--arch', default='mlp',
                        help='model architecture: ' + ' | '.join(MODEL_NAMES))

    # training and learning scheme.
    parser.add_argument('--stop_criteria', type=str, default='epoch')
    parser.add_argument('--num_epochs', type=int, default=None)
    parser.add_argument('--num_iterations', type=int, default=None)

    parser.add_argument('--local_step', type=int, default=1)
    parser.add_argument('--local_step_warmup_per_interval', default=False,
                        type=str2bool)
    parser.add_argument('--local_step_warmup_type', default=None, type=str)
    parser.add_argument('--local_step_warmup_period', default=None, type=int)
    parser.add_argument('--turn_on_local_step_from', default=None, type=int)
    parser.add_argument('--turn_off_local_step_from', default=None, type=int)

    parser.add_argument('--avg_model', type=str2bool, default=False)
    parser.add_argument('--reshuffle_per_epoch', default=False, type=str2bool)
    parser.add_argument('-b', '--batch_size', default=50, type=int)
    parser.add_argument('--growing_batch_size', default=False, type=str2bool)
    parser.add_argument('--base_batch_size', default=None, type=int)
    parser.add_argument('--max_batch_size', default=0, type=int)

    # learning rate scheme.
    parser.add_argument('--lr', type=float, default=0.01)
    parser.add_argument('--lr_schedule_scheme', type=str, default=None)
    parser.add_argument('--lr_change_epochs', type=str, default=None)
    parser.add_argument('--lr_fields', type=str, default=None)
    parser.add_argument('--lr_scale_indicators', type=str, default=None)
    parser.add_argument('--lr_scaleup', type=str2bool, default=False)
    parser.add_argument('--lr_scaleup_type', type=str, default='linear')
    parser.add_argument('--lr_scale_at_sync', type=float, default=1.0)
    parser.add_argument('--lr_warmup', type=str2bool, default=False)
    parser.add_argument('--lr_warmup_epochs', type=int, default=5)
    parser.add_argument('--lr_decay', type=float, default=10)
    parser.add_argument('--lr_onecycle_low', type=float, default=0.15)
    parser.add_argument('--lr_onecycle_high', type=float, default=3)
    parser.add_argument('--lr_onecycle_extra_low', type=float, default=0.0015)
    parser.add_argument('--lr_onecycle_num_epoch', type=int, default=46)
    parser.add_argument('--lr_gamma', type=float, default=None)
    parser.add_argument('--lr_mu', type=float, default=None)
    parser.add_argument('--lr_alpha', type=float, default=None)

    # optimizer.
    parser.add_argument('--optimizer', type=str, default='sgd')

    # momentum scheme.
    parser.add_argument('--in_momentum', type=str2bool, default=False)
    parser.add_argument('--in_momentum_factor', default=0.9, type=float)
    parser.add_argument('--out_momentum', type=str2bool, default=False)
    parser.add_argument('--out_momentum_factor', default=None, type=float)
    parser.add_argument('--use_nesterov', default=False, type=str2bool)

    # regularization.
    parser.add_argument('--weight_decay', default=5e-4, type=float)
    parser.add_argument('--correct_wd', type=str2bool, default=False)
    parser.add_argument('--drop_rate', default=0.0, type=float)

    # per-model parameters.
    parser.add_argument('--densenet_growth_rate', default=12, type=int)
    parser.add_argument('--densenet_bc_mode', default=False, type=str2bool)
    parser.add_argument('--densenet_compression', default=0.5, type=float)
    parser.add_argument('--wideresnet_widen_factor', default=4, type=int)
    parser.add_argument('--mlp_num_layers', default=2, type=int)
    parser.add_argument('--mlp_hidden_size', default=500, type=int)
    parser.add_argument('--rnn_seq_len', default=50, type=int)
    parser.add_argument('--rnn_hidden_size', default=50, type=int)
    parser.add_argument('--vocab_size', default=86, type=int)

    # misc.
    parser.add_argument('--manual_seed', type=int, default=6)
    parser.add_argument('-e', '--evaluate', dest='evaluate', type=str2bool,
                        default=False)
    parser.add_argument('--eval_freq', default=1, type=int)
    parser.add_argument('--summary_freq', default=10, type=int)
    parser.add_argument('--timestamp', default=None, type=str)

    # checkpoint.
    parser.add_argument('--debug', type=str2bool, default=False)
    parser.add_argument('--resume', default=None, type=str)
    parser.add_argument('--check_model_at_sync', default=False, type=str2bool)
    parser.add_argument('--track_model_aggregation', default=False,
                        type=str2bool)
    parser.add_argument('--checkpoint', '-c', default='./checkpoint/', type=str)
    parser.add_argument('--checkpoint_index', type=str, default=None)
    parser.add_argument('--save_all_models', type=str2bool, default=False)
    parser.add_argument('--save_some_models', type=str, default='1,29,59')
    parser.add_argument('--log_dir', default='./logdir/')
    parser.add_argument('--plot_dir', default=None, type=str)
    parser.add_argument('--pretrained', dest='pretrained', type=str2bool,
                        default=False)

    # device / topology.
    parser.add_argument('--is_distributed', default=True, type=str2bool)
    parser.add_argument('--experiment', type=str, default=None)
    parser.add_argument('--hostfile', type=str, default='hostfile')
    parser.add_argument('-j', '--num_workers', default=4, type=int)
    parser.add_argument('--dist_backend', default='nccl', type=str,
                        help='torch.distributed backend; "nccl" IS RCCL on '
                             'ROCm (GPU-direct over xGMI), "gloo" for CPU.')
    parser.add_argument('--blocks', default='2,2', type=str)
    parser.add_argument('--on_cuda', type=str2bool, default=True)
    parser.add_argument('--world', default=None, type=str)

    # --- MI355X-native additions (not in the reference) ---
    parser.add_argument('--bf16', type=str2bool, default=False,
                        help='bf16 autocast compute with fp32 master arena.')
    parser.add_argument('--hip_kernels', type=str2bool, default=True,
                        help='use the hand-written CDNA4 HIP kernel pack for '
                             'the arena hot paths (GPU only). When False, '
                             'fall back to eager torch ops.')
    parser.add_argument('--device_data_cache', type=str2bool, default=True,
                        help='keep each client partition resident in HBM '
                             'and serve batches as tensor slices with '
                             'batched on-GPU augmentation (GPU runs only)')
    parser.add_argument('--hip_graph', type=str2bool, default=False,
                        help='reserved. hipGraph capture of the local step '
                             'is implemented in the benchmark path '
                             '(bench.py --graph, two-capture stolen-grad '
                             'scheme); the parity training loops run eager '
                             'because per-step LR scheduling and per-step '
                             'console metrics are part of the reference '
                             'semantics a captured step would bake in.')
    parser.add_argument('--clients_per_rank', default=1, type=int,
                        help='virtual clients packed per GPU rank (their '
                             'replicas and aux state stay resident in HBM3E).')
    # NHWC is the native layout on MI355X: MIOpen's igemm solvers are NHWC
    # (NCHW tensors pay batched_transpose wrappers around every conv) and
    # the fused BN / stem kernels have NHWC paths (profiles/r01_bench_notes.md)
    parser.add_argument('--channels_last', type=str2bool, default=True)
    parser.add_argument('--aggregate_bn_stats', type=str2bool, default=True,
                        help='average BatchNorm running stats across online '
                             'clients at sync (the reference never syncs '
                             'buffers, leaving the server model with init '
                             'stats).')
    parser.add_argument('--fused_bn', type=str2bool, default=True,
                        help='replace nn.BatchNorm2d with the fused gfx950 '
                             'BN kernels for GPU training.')
    return parser


def derive_args(args):
    """Post-parse validation/derivation (reference `parameters.py:239-259`)."""
    if args.timestamp is None:
        # import here to avoid a cycle at module import time.
        from fedtorch_amd.logs.checkpoint import get_checkpoint_folder_name
        args.timestamp = get_checkpoint_folder_name(args)
    if args.growing_batch_size and args.base_batch_size is None:
        args.base_batch_size = 1
    if args.federated:
        if args.reshuffle_per_epoch:
            raise ValueError(
                'In the Federated Learning mode data cannot be reshuffled in '
                'the middle of training; set --reshuffle_per_epoch False')
        args.num_epochs = int(
            args.num_epochs_per_comm * args.num_comms * args.online_client_rate)
        # keep at least one epoch so stop criteria / schedulers are sane.
        args.num_epochs = max(args.num_epochs, 1)
        if args.federated_type == 'afl':
            args.federated_sync_type = 'local_step'
            args.local_step = 1
        if args.federated_type == 'qsparse':
            # NOTE: the reference has `args.compressed == True` (a no-op
            # comparison, `parameters.py:253`); qsparse REQUIRES compression,
            # so we actually set it.
            args.compressed = True
        if args.quantized and args.compressed and args.federated_type != 'qsparse':
            raise ValueError('Quantization is mutually exclusive with '
                             'compression; choose only one.')
        if args.federated_type in ('apfl', 'perfedme', 'perfedavg'):
            args.fed_personal = True
    if args.num_epochs is None:
        args.num_epochs = 1
    return args


def get_args(arg_list=None):
    """Parse known args (reference `parameters.py:12` — public API)."""
    parser = build_parser()
    args = parser.parse_args(arg_list)
    return derive_args(args)


def print_args(args):
    print('parameters: ')
    for arg in vars(args):
        print(arg, getattr(args, arg))
