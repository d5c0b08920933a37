# -*- coding: utf-8 -*-
"""Closed-form oracles for the math hot paths (CPU/eager path; the GPU
kernels are compared against these in tests/test_gpu_kernels.py)."""
import numpy as np
import pytest
import torch

from fedtorch_amd import ops


def reference_quantize(x, num_bits=8):
    """Literal re-statement of reference `flow_utils.py:169-205`."""
    qmin = -2.0 ** (num_bits - 1)
    qmax = 2.0 ** (num_bits - 1) - 1.0
    min_val, max_val, mean_val = x.min(), x.max(), x.mean()
    scale = (max_val - min_val) / (qmax - qmin)
    if scale == 0.0:
        scale = 0.001
    initial_zero_point = qmin - (min_val - mean_val) / scale
    if initial_zero_point < qmin:
        zero_point = qmin
    elif initial_zero_point > qmax:
        zero_point = qmax
    else:
        zero_point = int(initial_zero_point)
    q_x = zero_point + (x - mean_val) / scale
    q_x.clamp_(qmin, qmax).round_()
    q_x = q_x.round().char() if num_bits == 8 else q_x.round().short()
    return q_x, torch.tensor([float(scale), float(zero_point),
                              float(mean_val)])


@pytest.mark.parametrize('bits', [8, 16])
def test_quantize_matches_reference(bits):
    torch.manual_seed(0)
    x = torch.randn(10000)
    q, info = ops.quantize(x, bits)
    qr, infor = reference_quantize(x.clone(), bits)
    assert torch.equal(q, qr)
    assert torch.allclose(info.cpu(), infor, atol=1e-5)


def test_quantize_zero_tensor():
    x = torch.zeros(256)
    q, info = ops.quantize(x, 8)
    xr = ops.dequantize(q, info)
    assert torch.allclose(xr, x)  # zeros dequantize to exactly zero


def test_quantize_roundtrip_error_bound():
    torch.manual_seed(1)
    x = torch.randn(4096)
    q, info = ops.quantize(x, 8)
    xr = ops.dequantize(q, info)
    # scale/2 quantization + up to 1 grid unit from zero-point truncation
    assert (x - xr).abs().max() <= info[0].item() * 1.5 + 1e-6


def test_dequant_accumulate():
    torch.manual_seed(2)
    xs = [torch.randn(512) for _ in range(4)]
    qs, infos = zip(*[ops.quantize(x, 8) for x in xs])
    out = torch.zeros(512)
    ops.dequant_accumulate(torch.stack(qs), torch.stack(infos), out)
    expected = sum(ops.dequantize(q, i) for q, i in zip(qs, infos))
    assert torch.allclose(out, expected, atol=1e-5)


def test_topk_matches_torch():
    torch.manual_seed(3)
    x = torch.randn(1000)
    v, i = ops.topk_compress(x, 100)
    vr, ir = x.abs().topk(100)
    assert set(i.tolist()) == set(ir.tolist())
    assert torch.allclose(v.abs().sort()[0], vr.sort()[0])
    # values carry the ORIGINAL signs (reference `flow_utils.py:228` v=x[i])
    assert torch.equal(v, x[i.long()])


def test_scatter_accumulate():
    torch.manual_seed(4)
    n, k = 256, 32
    xs = [torch.randn(n) for _ in range(3)]
    vs, idxs = [], []
    for x in xs:
        v, i = ops.topk_compress(x, k)
        vs.append(v)
        idxs.append(i)
    out = torch.zeros(n)
    ops.scatter_accumulate(out, torch.stack(vs), torch.stack(idxs))
    expected = torch.zeros(n)
    for v, i in zip(vs, idxs):
        expected.scatter_add_(0, i.long(), v)
    assert torch.allclose(out, expected)


def test_error_feedback_and_delta():
    torch.manual_seed(5)
    mem = torch.randn(64)
    mem0 = mem.clone()
    g = torch.randn(64)
    d = torch.randn(64)
    ops.error_feedback_update(mem, g, d, 4.0)
    assert torch.allclose(mem, mem0 + 4.0 * g - d)

    delta = torch.randn(64)
    d0 = delta.clone()
    s, agg, c = torch.randn(64), torch.randn(64), torch.randn(64)
    ops.delta_update(delta, s, agg, c, 0.5)
    assert torch.allclose(delta, d0 + 0.5 * (s - agg - c))


def test_scaffold_control_update():
    torch.manual_seed(6)
    cc, cs, s, c = [torch.randn(64) for _ in range(4)]
    out = torch.zeros(64)
    ops.scaffold_control_update(out, cc, cs, s, c, 2.0)
    assert torch.allclose(out, cc - cs + 2.0 * (s - c))


def test_weighted_diff_restore():
    s = torch.randn(64)
    c = torch.randn(64)
    out = torch.zeros(64)
    c0 = c.clone()
    ops.weighted_diff_restore(s, c, out, 0.25)
    assert torch.allclose(out, (s - c0) * 0.25)
    assert torch.equal(c, s)


def test_simplex_projection():
    torch.manual_seed(7)
    v = torch.randn(50)
    w = ops.euclidean_proj_simplex(v.clone())
    # numpy sorted-reference (reference `flow_utils.py:141-157`)
    vn = v.numpy().astype(np.float64)
    u = np.sort(vn)[::-1]
    cssv = np.cumsum(u) - 1.0
    ind = np.arange(50) + 1
    cond = u - cssv / ind > 0
    rho = ind[cond][-1]
    theta = cssv[cond][-1] / float(rho)
    wr = np.maximum(vn - theta, 0)
    assert np.allclose(w.numpy(), wr, atol=1e-5)
    assert abs(w.sum().item() - 1.0) < 1e-5
    assert (w >= 0).all()


def test_blend_and_alpha_grad():
    a, b = torch.randn(64), torch.randn(64)
    out = torch.zeros(64)
    ops.blend(out, a, b, 0.3)
    assert torch.allclose(out, 0.3 * a + 0.7 * b, atol=1e-6)

    lf, pf, lg, pg = [torch.randn(128) for _ in range(4)]
    ga = ops.alpha_grad(lf, pf, lg, pg, 0.4)
    expected = torch.dot(pf - lf, 0.4 * pg + 0.6 * lg).item() + 0.02 * 0.4
    assert abs(ga - expected) < 1e-4


def test_rank_weight_semantics():
    """The weighting that makes the all-reduce identical to the reference's
    star gather+sum+broadcast (`fedavg.py:17-27`, `qsparse.py:23`)."""
    from types import SimpleNamespace
    from fedtorch_amd.aggregation.federated import rank_weight

    def args_for(rank, n=4):
        return SimpleNamespace(graph=SimpleNamespace(rank=rank, n_nodes=n),
                               num_samples_per_epoch=100,
                               train_dataset_size=400)

    # online clients include the server: uniform 1/K
    w, k, part = rank_weight(args_for(1), [0, 1, 2])
    assert (w, k, part) == (1.0 / 3, 3, True)
    # server offline: counted in K but contributes 0 (`fedavg.py:19-20`)
    w, k, part = rank_weight(args_for(0), [1, 2])
    assert (w, k, part) == (0.0, 3, False)
    w, k, part = rank_weight(args_for(1), [1, 2])
    assert (w, k, part) == (1.0 / 3, 3, True)
    # fully offline rank never contributes
    w, _, part = rank_weight(args_for(3), [1, 2])
    assert (w, part) == (0.0, False)
    # DRFA/AFL: lambda_i * n / K (`fedavg.py:27`)
    w, _, _ = rank_weight(args_for(2), [0, 1, 2], lambda_weight=0.5)
    assert abs(w - 0.5 * 4 / 3) < 1e-9
    # qsparse: sample-proportional (`qsparse.py:23`)
    w, _, _ = rank_weight(args_for(2), [0, 1, 2], sample_proportional=True)
    assert abs(w - 0.25) < 1e-9


def test_fedadam_normalize_math():
    """FedAdam server normalizer (the reference's `fedavg.py:81-85` crashes
    on a missing np import; this pins our working version)."""
    import numpy as np
    from types import SimpleNamespace
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    from fedtorch_amd.aggregation.federated import _fedadam_normalize
    m = nn.Linear(4, 2)
    a = Arena(m)
    agg = torch.ones_like(a.flat)
    args = SimpleNamespace(fedadam_beta=0.9, fedadam_tau=0.1,
                           fedadam_v=[1.0, 1.0])
    views = a.views_of(agg)
    norms = [float(torch.norm(v)) for v in views]
    _fedadam_normalize(args, a, agg)
    for i, v in enumerate(a.views_of(agg)):
        v_exp = 0.9 * 1.0 + 0.1 * norms[i]
        assert abs(args.fedadam_v[i] - v_exp) < 1e-6
        assert torch.allclose(v, torch.ones_like(v) /
                              (np.sqrt(v_exp) + 0.1), atol=1e-6)


def test_inference_personal_blend():
    """alpha-blend of two models' logits (reference `eval.py:31-39`)."""
    import torch.nn as nn
    from fedtorch_amd.trainings.eval import inference_personal
    torch.manual_seed(5)
    m1, m2 = nn.Linear(6, 3), nn.Linear(6, 3)
    x = torch.randn(10, 6)
    y = torch.randint(0, 3, (10,))
    crit = nn.CrossEntropyLoss()
    loss, perf = inference_personal(m1, m2, 0.3, crit, (1,), x, y)
    expected = crit(0.3 * m1(x) + 0.7 * m2(x), y)
    assert torch.allclose(loss, expected, atol=1e-6)
    assert 0.0 <= perf[0] <= 100.0
