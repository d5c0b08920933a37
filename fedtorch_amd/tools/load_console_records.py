# -*- coding: utf-8 -*-
"""Offline parsing of per-rank `record{rank}` console logs into pandas
frames (parity with reference `fedtorch/tools/load_console_records.py`:
same line formats, same series names)."""
import os
import re
from datetime import datetime

import pandas as pd

TIME_FMT = '%Y:%m:%d %H:%M:%S'

PAT_TRAIN = (r'(.*?)\tProcess: \d+: Epoch: (.*?)\. Local index: (.*?)\. '
             r'Load: (.*?)s \| Data: (.*?)s \| Computing: (.*?)s \| '
             r'Sync: (.*?)s \| Global: (.*?)s \| Loss: (.*?) \| '
             r'top1: (.*?) \| top5: (.*?) \| learning_rate: (.*?) \| '
             r'rounds_comm: (.*)')
VAR_TRAIN = ['time', 'epoch', 'local_index', 'load_time', 'data_time',
             'compute_time', 'sync_time', 'global_time', 'loss', 'top1',
             'top5', 'learning_rate', 'rounds_comm']

PAT_TEST = (r'(.*?)\tTest at batch: (.*?)\. Epoch: (.*?)\. Process: (.*?)\. '
            r'Prec@1: (.*?) Prec@5: (.*?) Loss: (.*?) Comm: (.*)')
VAR_TEST = ['time', 'batch', 'epoch', 'Process', 'top1', 'top5', 'loss',
            'comm']

PAT_VAL = (r'(.*?)\t(Personal|Global) performance for (validation|train) at '
           r'batch: (.*?)\. Epoch: (.*?)\. Process: (.*?)\. Prec@1: (.*?) '
           r'Prec@5: (.*?) Loss: (.*?) Comm: (.*)')
VAR_VAL = ['time', 'mode', 'split', 'batch', 'epoch', 'Process', 'top1',
           'top5', 'loss', 'comm']

PAT_COMM = r'(.*?)\tThis round communication time is: (.*)'
VAR_COMM = ['time', 'comm_time']

PAT_STAT = (r'(.*?)\t(Personal|Global) per client stat for '
            r'(validation|train) at batch: (.*?)\. Epoch: (.*?)\. '
            r'Process: (.*?)\. Worst: (.*?) Best: (.*?) Var: (.*?) Comm: (.*)')
VAR_STAT = ['time', 'mode', 'split', 'batch', 'epoch', 'Process', 'worst',
            'best', 'var', 'comm']


def _to_float(x):
    try:
        return float(x)
    except (TypeError, ValueError):
        return x


def _parse(lines, pattern, var_names):
    rows = []
    t0 = None
    for line in lines:
        m = re.findall(pattern, line, re.DOTALL)
        if not m:
            continue
        vals = [x.strip() if isinstance(x, str) else x for x in m[0]]
        t = datetime.strptime(vals[0], TIME_FMT)
        if t0 is None:
            t0 = t
        vals[0] = (t - t0).total_seconds()
        rows.append([_to_float(v) for v in vals])
    return pd.DataFrame(rows, columns=var_names)


def read_record(path):
    with open(path) as f:
        return f.read().splitlines()


def parse_record_for_train(path):
    return _parse(read_record(path), PAT_TRAIN, VAR_TRAIN)


def parse_record_for_test(path):
    return _parse(read_record(path), PAT_TEST, VAR_TEST)


def parse_record_for_val(path):
    return _parse(read_record(path), PAT_VAL, VAR_VAL)


def parse_record_for_comm_time(path):
    return _parse(read_record(path), PAT_COMM, VAR_COMM)


def parse_record_for_per_client_stat(path):
    return _parse(read_record(path), PAT_STAT, VAR_STAT)


def save_records_to_csv(ckpt_dir, num_workers=1, dest_dir=None,
                        test_save=True, val_save=True, comm_time=True,
                        train_save=True):
    """Parse every rank's record file under a run dir into CSVs
    (reference `load_console_records.py:222-274`)."""
    dest_dir = dest_dir or ckpt_dir
    os.makedirs(dest_dir, exist_ok=True)
    for rank in range(num_workers):
        rec = os.path.join(ckpt_dir, str(rank), 'record%d' % rank)
        if not os.path.exists(rec):
            continue
        if train_save:
            parse_record_for_train(rec).to_csv(
                os.path.join(dest_dir, 'train_%d.csv' % rank), index=False)
        if test_save:
            parse_record_for_test(rec).to_csv(
                os.path.join(dest_dir, 'test_%d.csv' % rank), index=False)
        if val_save:
            parse_record_for_val(rec).to_csv(
                os.path.join(dest_dir, 'val_%d.csv' % rank), index=False)
        if comm_time:
            parse_record_for_comm_time(rec).to_csv(
                os.path.join(dest_dir, 'comm_%d.csv' % rank), index=False)


def get_checkpoint_args(path):
    import torch
    ckpt = torch.load(path, map_location='cpu', weights_only=False)
    return ckpt['arguments']
