# -*- coding: utf-8 -*-
"""Aux helpers (parity with reference `fedtorch/utils/auxiliary.py`)."""
from copy import deepcopy


def deepcopy_model(args, model):
    """Deep-copy a model; RNN hidden state must not be copied
    (reference keeps persistent hidden state on the live model only,
    `fedtorch/models/nonconvex/rnn.py:27-35`)."""
    if getattr(args, 'arch', '') == 'rnn' and hasattr(model, 'hidden'):
        hidden = model.hidden
        model.hidden = None
        out = deepcopy(model)
        model.hidden = hidden
        return out
    return deepcopy(model)


def dict2obj(d):
    class _Obj(object):
        pass
    o = _Obj()
    for k, v in d.items():
        setattr(o, k, dict2obj(v) if isinstance(v, dict) else v)
    return o
