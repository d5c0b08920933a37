# -*- coding: utf-8 -*-
"""Criterion factory (parity with reference `components/criterion.py:6-11`)."""
import torch.nn as nn


def define_criterion(args):
    if 'least_square' in args.arch:
        return nn.MSELoss(reduction='mean')
    return nn.CrossEntropyLoss(reduction='mean')
