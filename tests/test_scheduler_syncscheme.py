# -*- coding: utf-8 -*-
"""LR schedule curves + sync-scheme list vs closed forms."""
import types

from fedtorch_amd.components.scheduler import define_lr_scheduler
from fedtorch_amd.aggregation.distributed import define_sync_freq


def args_ns(**kw):
    base = dict(
        lr=0.1, batch_size=50, lr_scaleup=False, lr_scaleup_type='linear',
        base_batch_size=None, lr_schedule_scheme=None, lr_change_epochs=None,
        lr_fields=None, lr_scale_indicators=None, lr_warmup=False,
        lr_warmup_epochs=5, lr_decay=10.0, lr_onecycle_low=0.15,
        lr_onecycle_high=3.0, lr_onecycle_extra_low=0.0015,
        lr_onecycle_num_epoch=46, lr_gamma=None, lr_mu=None, lr_alpha=None,
        num_epochs=100)
    base.update(kw)
    ns = types.SimpleNamespace(**base)
    ns.graph = types.SimpleNamespace(n_nodes=4)
    return ns


def test_multistep_decay():
    a = args_ns(lr_schedule_scheme='custom_multistep',
                lr_change_epochs='20,40')
    f = define_lr_scheduler(a)
    assert abs(f(0.0) - 0.1) < 1e-9
    assert abs(f(19.99) - 0.1) < 1e-9
    assert abs(f(20.0) - 0.01) < 1e-9
    assert abs(f(45.0) - 0.001) < 1e-9


def test_convex_decay():
    a = args_ns(lr_schedule_scheme='custom_convex_decay', lr_gamma=1.0,
                lr_mu=2.0, lr_alpha=1.0, num_epochs=10)
    f = define_lr_scheduler(a)
    # gamma / (mu (t + a))
    assert abs(f(0.0) - 1.0 / 2.0) < 1e-9
    assert abs(f(3.0) - 1.0 / (2 * 4.0)) < 1e-9


def test_onecycle_shape():
    a = args_ns(lr_schedule_scheme='custom_one_cycle', num_epochs=50)
    f = define_lr_scheduler(a)
    assert abs(f(0.0) - 0.15) < 1e-9
    assert f(10.0) > f(0.0)  # rising phase
    assert abs(f(23.0) - 3.0) < 0.15  # near peak at half-cycle
    assert f(47.0) < 0.15  # extra-low tail


def test_linear_scaleup():
    a = args_ns(lr_scaleup=True)
    define_lr_scheduler(a)
    assert abs(a.learning_rate - 0.1 * 4) < 1e-9


def test_sync_freq_basic():
    steps = define_sync_freq(
        num_epochs=10, local_step=4, local_step_warmup_type=None,
        local_step_warmup_period=None, turn_on_local_step_from=None,
        turn_off_local_step_from=None, warmup_per_intervals=False,
        lr_change_epochs=None)
    assert len(steps) == 12  # num_epochs + 2
    assert all(s == 4 for s in steps)


def test_sync_freq_linear_warmup():
    steps = define_sync_freq(
        num_epochs=10, local_step=8, local_step_warmup_type='linear',
        local_step_warmup_period=4, turn_on_local_step_from=None,
        turn_off_local_step_from=None, warmup_per_intervals=False,
        lr_change_epochs=None)
    assert steps[:4] == [2, 4, 6, 8]
    assert all(s == 8 for s in steps[4:])


def test_sync_freq_turn_off():
    steps = define_sync_freq(
        num_epochs=10, local_step=4, local_step_warmup_type=None,
        local_step_warmup_period=None, turn_on_local_step_from=None,
        turn_off_local_step_from=5, warmup_per_intervals=False,
        lr_change_epochs='5')
    assert all(s == 4 for s in steps[:5])
    assert all(s == 1 for s in steps[5:])


def test_parameter_derivations_match_reference():
    """Post-parse derivations (reference `parameters.py:240-260`)."""
    import pytest as _pytest
    from fedtorch_amd.parameters import get_args
    base = ['-d', 'mnist', '-a', 'mlp', '-f', 'true', '--num_comms', '10',
            '--num_epochs_per_comm', '2', '--online_client_rate', '0.5',
            '--checkpoint', '/tmp/ft_deriv', '--debug', 'false']
    a = get_args(base + ['--federated_type', 'fedavg'])
    # num_epochs = per_comm * comms * online_rate (`:248`)
    assert a.num_epochs == int(2 * 10 * 0.5)
    # afl forces local_step sync with step 1 (`:250-252`)
    a = get_args(base + ['--federated_type', 'afl'])
    assert a.federated_sync_type == 'local_step' and a.local_step == 1
    # personalization forces fed_personal (`:257-259`)
    for t in ('apfl', 'perfedme', 'perfedavg'):
        a = get_args(base + ['--federated_type', t])
        assert a.fed_personal, t
    # qsparse implies compression (the reference's `:253` is a no-op bug —
    # fixed here, documented in parameters.py)
    a = get_args(base + ['--federated_type', 'qsparse'])
    assert a.compressed
    # quantize + compress are mutually exclusive (`:254-255`)
    with _pytest.raises(ValueError):
        get_args(base + ['--federated_type', 'fedavg', '--quantized', 'true',
                         '--compressed', 'true'])
    # reshuffle is rejected under federation (`:246-247`)
    with _pytest.raises(ValueError):
        get_args(base + ['--federated_type', 'fedavg',
                         '--reshuffle_per_epoch', 'true'])
