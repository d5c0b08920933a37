# -*- coding: utf-8 -*-
"""Model zoo (parity with reference `fedtorch/components/models/`).

Factories all take ``args`` and dispatch on ``args.arch`` / ``args.data``.
"""
from fedtorch_amd.components.models.convex import (  # noqa: F401
    logistic_regression, least_square,
    robust_logistic_regression, robust_least_square)
from fedtorch_amd.components.models.mlp import mlp, robust_mlp  # noqa: F401
from fedtorch_amd.components.models.cnn import cnn  # noqa: F401
from fedtorch_amd.components.models.rnn import rnn  # noqa: F401
from fedtorch_amd.components.models.resnet import resnet  # noqa: F401
from fedtorch_amd.components.models.densenet import densenet  # noqa: F401
from fedtorch_amd.components.models.wideresnet import wideresnet  # noqa: F401
