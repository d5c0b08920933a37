# -*- coding: utf-8 -*-
"""Tiny file helpers (parity with reference `fedtorch/utils/op_files.py`)."""
import json
import os
import pickle


def read_txt(path):
    with open(path, 'r') as f:
        return f.read().splitlines()


def write_txt(data, out_path, type="w"):
    with open(out_path, type) as f:
        f.write(data)


def load_pickle(path):
    with open(path, 'rb') as f:
        return pickle.load(f)


def write_pickle(data, path):
    os.makedirs(os.path.dirname(path) or '.', exist_ok=True)
    with open(path, 'wb') as f:
        pickle.dump(data, f)


def load_json(path):
    with open(path, 'r') as f:
        return json.load(f)


def write_json(data, path):
    os.makedirs(os.path.dirname(path) or '.', exist_ok=True)
    with open(path, 'w') as f:
        json.dump(data, f, indent=2)
