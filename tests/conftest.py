import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault('FEDTORCH_SYNTH_SIZE', '512')


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs an MI355X (or any ROCm GPU); run with -m gpu')


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason='no GPU in this environment')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def tiny_args():
    """Small CPU config for unit tests."""
    from fedtorch_amd.parameters import get_args
    args = get_args([
        '-d', 'mnist', '-a', 'logistic_regression', '-f', 'true',
        '--federated_type', 'fedavg', '--num_comms', '2',
        '--online_client_rate', '1.0', '-b', '25', '--lr', '0.1',
        '--on_cuda', 'false', '-j', '0', '--checkpoint', '/tmp/ft_test_ckpt',
        '--debug', 'false'])
    return args
