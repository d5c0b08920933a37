# -*- coding: utf-8 -*-
"""Path helpers (parity with reference `fedtorch/utils/op_paths.py`)."""
import os
import shutil


def build_dirs(path):
    os.makedirs(path, exist_ok=True)


def remove_folder(path):
    if os.path.isdir(path):
        shutil.rmtree(path, ignore_errors=True)


def list_files(root):
    out = []
    for dirpath, _, filenames in os.walk(root):
        for fn in filenames:
            out.append(os.path.join(dirpath, fn))
    return out
