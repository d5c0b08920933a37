# -*- coding: utf-8 -*-
"""Run summarizer (parity with reference `fedtorch/tools/get_summary.py`):
walk a checkpoint tree, pull hyperparameters + best accuracies."""
import os

import pandas as pd

from fedtorch_amd.tools.load_console_records import (
    parse_record_for_test, parse_record_for_comm_time)

INTERESTED_ARGS = ['data', 'arch', 'federated_type', 'lr', 'batch_size',
                   'num_comms', 'num_epochs_per_comm', 'online_client_rate',
                   'local_step', 'weight_decay', 'quantized', 'compressed']


def summarize_run(run_dir):
    """One run dir ({ckpt}/{data}/{arch}/{exp}/{timestamp}) -> dict."""
    out = {'run': os.path.basename(run_dir)}
    rec0 = os.path.join(run_dir, '0', 'record0')
    if os.path.exists(rec0):
        test = parse_record_for_test(rec0)
        if len(test):
            out['best_top1'] = float(test['top1'].max())
            out['final_top1'] = float(test['top1'].iloc[-1])
            out['rounds'] = int(test['comm'].iloc[-1])
        comm = parse_record_for_comm_time(rec0)
        if len(comm):
            out['total_comm_time'] = float(comm['comm_time'].sum())
    ckpt = os.path.join(run_dir, 'checkpoint.pth.tar')
    if os.path.exists(ckpt):
        from fedtorch_amd.tools.load_console_records import \
            get_checkpoint_args
        args = get_checkpoint_args(ckpt)
        for k in INTERESTED_ARGS:
            out[k] = getattr(args, k, None)
    return out


def summarize_tree(root):
    rows = []
    for dirpath, dirnames, filenames in os.walk(root):
        if any(d.isdigit() for d in dirnames) and \
                any(f.startswith('record') or f == 'checkpoint.pth.tar'
                    for d in dirnames
                    for f in os.listdir(os.path.join(dirpath, d))
                    if os.path.isdir(os.path.join(dirpath, d))):
            rows.append(summarize_run(dirpath))
            dirnames[:] = []
    return pd.DataFrame(rows)


if __name__ == '__main__':
    import sys
    print(summarize_tree(sys.argv[1] if len(sys.argv) > 1
                         else './checkpoint'))
