# -*- coding: utf-8 -*-
"""Launcher (parity with reference `run_mpi.py`, same short flags): builds a
torchrun command — one process per MI355X GPU over RCCL — instead of
`mpirun --oversubscribe`.

    python run_dist.py -f -ft fedavg -n 8 -d cifar10 -b 128 -oc
"""
import argparse
import os
import shlex
import subprocess
import sys

DEFAULT_MODEL = {'epsilon': 'logistic_regression', 'MSD': 'robust_least_square',
                 'cifar10': 'logistic_regression', 'emnist': 'mlp',
                 'emnist_full': 'mlp', 'mnist': 'mlp',
                 'synthetic': 'logistic_regression', 'fashion_mnist': 'mlp',
                 'adult': 'logistic_regression'}
MLP_SIZE = {'mnist': 200, 'fashion_mnist': 200, 'cifar10': 200,
            'cifar100': 500, 'adult': 50, 'MSD': 50, 'emnist': 200,
            'emnist_full': 200}


def main(args):
    if getattr(args, 'tmp_dir', None):
        os.environ['TMPDIR'] = args.tmp_dir
    blocks = str(args.num_clients)
    world = ','.join(str(x) for x in range(args.num_clients))
    params = {
        '--avg_model': True,
        '--debug': True,
        '--eval_freq': 1,
        '--stop_criteria': 'epoch',
        '--num_epochs': args.num_epochs_per_comm * args.num_comms,
        '--on_cuda': args.on_cuda,
        '--num_workers': 0,
        '--blocks': blocks,
        '--world': world,
        '--weight_decay': args.weight_decay,
        '--use_nesterov': False,
        '--in_momentum': False,
        '--out_momentum': False,
        '--local_step': args.local_steps,
        '--turn_on_local_step_from': 0,
        '--checkpoint': args.data_path,
        '--drop_rate': 0.25,
        '--arch': args.arch or DEFAULT_MODEL.get(args.dataset, 'mlp'),
        '--mlp_num_layers': 2,
        '--mlp_hidden_size': MLP_SIZE.get(args.dataset, 200),
        '--data': args.dataset,
        '--data_dir': args.data_path,
        '--synthetic_alpha': args.synthetic_params[0],
        '--synthetic_beta': args.synthetic_params[1],
        '--batch_size': args.batch_size,
        '--partition_data': True,
        '--reshuffle_per_epoch': not args.federated,
        '--iid_data': args.iid,
        '--num_class_per_client': args.num_class_per_client,
        '--unbalanced': args.unbalanced,
        '--federated': args.federated,
        '--federated_type': args.federated_type,
        '--federated_sync_type': args.federated_sync_type,
        '--num_comms': args.num_comms,
        '--online_client_rate': args.online_client_rate,
        '--num_epochs_per_comm': args.num_epochs_per_comm,
        '--fed_personal': args.fed_personal,
        '--quantized': args.quantized,
        '--quantized_bits': args.quantized_bits,
        '--compressed': args.compressed,
        '--compressed_ratio': args.compressed_ratio,
        '--federated_drfa': args.federated_drfa,
        '--drfa_gamma': args.drfa_gamma,
        '--fed_adaptive_alpha': args.fed_adaptive_alpha,
        '--fed_personal_alpha': args.fed_personal_alpha,
        '--fedprox_mu': args.fedprox_mu,
        '--perfedavg_beta': 0.03,
        '--sensitive_feature': args.sensitive_feature,
        '--lr_schedule_scheme': 'custom_multistep',
        '--lr_change_epochs': ','.join(
            str(x) for x in
            range(1, args.num_epochs_per_comm * args.num_comms)),
        '--lr_warmup': False,
        '--lr': args.lr_gamma,
        '--lr_scale_at_sync': args.lr_sync,
        '--lr_warmup_epochs': 3,
        '--lr_decay': 1.01,
        '--bf16': args.bf16,
        '--hip_graph': getattr(args, 'hip_graph', False),
        '--channels_last': args.bf16,  # NHWC pairs with the bf16 path
    }
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', str(args.num_clients),
           '--master-addr', '127.0.0.1', '--master-port',
           str(args.master_port), '-m', 'fedtorch_amd.main']
    for k, v in params.items():
        if v is not None:
            cmd += [k, str(v)]
    print('\nRunning:\n' + ' '.join(shlex.quote(c) for c in cmd))
    if args.dry_run:
        return 0
    env = dict(os.environ)
    env.setdefault('HSA_ENABLE_IPC_MODE_LEGACY', '0')
    return subprocess.call(cmd, env=env)


if __name__ == '__main__':
    parser = argparse.ArgumentParser(
        description='Run fedtorch_amd with torchrun over RCCL/xGMI.')
    parser.add_argument('-e', '--num_epochs_per_comm', default=1, type=int)
    parser.add_argument('-n', '--num_clients', default=8, type=int)
    parser.add_argument('-d', '--dataset', default='mnist', type=str)
    parser.add_argument('-p', '--data_path', default='./data', type=str)
    parser.add_argument('-b', '--batch_size', default=50, type=int)
    parser.add_argument('-c', '--num_comms', default=100, type=int)
    parser.add_argument('-lg', '--lr_gamma', default=1.0, type=float)
    parser.add_argument('-lm', '--lr_mu', default=1, type=float)
    parser.add_argument('-ls', '--lr_sync', default=1.0, type=float)
    parser.add_argument('-w', '--weight_decay', default=1e-4, type=float)
    parser.add_argument('-i', '--iid', action='store_true')
    parser.add_argument('-l', '--local_steps', default=1, type=int)
    parser.add_argument('-oc', '--on_cuda', action='store_true')
    parser.add_argument('-f', '--federated', action='store_true')
    parser.add_argument('-ft', '--federated_type', default='fedavg', type=str)
    parser.add_argument('-fd', '--federated_drfa', action='store_true')
    parser.add_argument('-dg', '--drfa_gamma', default=0.1, type=float)
    parser.add_argument('-fs', '--federated_sync_type', default='epoch',
                        type=str, choices=['epoch', 'local_step'])
    parser.add_argument('-k', '--online_client_rate', default=1.0, type=float)
    parser.add_argument('-r', '--num_class_per_client', default=2, type=int)
    parser.add_argument('-sp', '--synthetic_params', nargs='+', type=float,
                        default=[0.0, 0.0])
    parser.add_argument('-q', '--quantized', action='store_true')
    parser.add_argument('-cp', '--compressed', action='store_true')
    parser.add_argument('-cr', '--compressed_ratio', default=1.0, type=float)
    parser.add_argument('-u', '--unbalanced', action='store_true')
    parser.add_argument('-fp', '--fed_personal', action='store_true')
    parser.add_argument('-pa', '--fed_personal_alpha', default=0.0,
                        type=float)
    parser.add_argument('-pd', '--fed_adaptive_alpha', action='store_true')
    parser.add_argument('-sf', '--sensitive_feature', default=9, type=int)
    parser.add_argument('-B', '--quantized_bits', default=8, type=int)
    parser.add_argument('-pm', '--fedprox_mu', default=0.002, type=float)
    parser.add_argument('-a', '--arch', default=None, type=str)
    # reference `run_mpi.py:140`: TMPDIR override for dataset staging
    parser.add_argument('-td', '--tmp_dir', default='/tmp', type=str)
    parser.add_argument('--bf16', action='store_true')
    parser.add_argument('--hip_graph', action='store_true',
                        help='hipGraph-capture the local steps '
                             '(trainings/graphstep.py)')
    parser.add_argument('--master_port', default=29500, type=int)
    parser.add_argument('--dry_run', action='store_true',
                        help='print the torchrun command and exit')
    sys.exit(main(parser.parse_args()))
