# -*- coding: utf-8 -*-
"""Property-based checks (hypothesis) over the eager math paths — the same
formulas the HIP kernels are pinned against in tests/test_gpu_kernels.py."""
import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

import fedtorch_amd.ops as ops


def _force_eager():
    old = ops.FORCE_EAGER
    ops.FORCE_EAGER = True
    return old


@settings(max_examples=30, deadline=None)
@given(st.integers(min_value=2, max_value=2048),
       st.integers(min_value=0, max_value=2 ** 31 - 1),
       st.sampled_from([8, 16]))
def test_quantize_roundtrip_bound_property(n, seed, bits):
    """|x - dequant(quant(x))| <= scale * 1.5 for any tensor (the 0.5 grid
    rounding plus <=1 grid unit from the reference's truncated zero-point,
    `flow_utils.py:183-192`)."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, generator=g) * (1 + seed % 7)
    old = _force_eager()
    try:
        q, info = ops.quantize(x, bits)
        xr = ops.dequantize(q, info)
    finally:
        ops.FORCE_EAGER = old
    scale = float(info[0])
    assert (x - xr).abs().max().item() <= scale * 1.5 + 1e-6


@settings(max_examples=30, deadline=None)
@given(st.integers(min_value=1, max_value=512),
       st.integers(min_value=0, max_value=2 ** 31 - 1))
def test_simplex_projection_properties(n, seed):
    """Projection lands on the simplex and is idempotent
    (`flow_utils.py:52-97`)."""
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(n, generator=g) * 3
    p = ops.euclidean_proj_simplex(v.clone())
    assert abs(float(p.sum()) - 1.0) < 1e-4
    assert float(p.min()) >= -1e-7
    p2 = ops.euclidean_proj_simplex(p.clone())
    assert torch.allclose(p, p2, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(st.integers(min_value=16, max_value=4096),
       st.integers(min_value=0, max_value=2 ** 31 - 1),
       st.floats(min_value=0.05, max_value=1.0))
def test_topk_budget_and_exactness_property(n, seed, ratio):
    """top-k keeps exactly k entries and they are the k largest |x|
    (`flow_utils.py:218-230`)."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, generator=g)
    k = max(int(n * ratio / 2), 1)
    old = _force_eager()
    try:
        v, i = ops.topk_compress(x, k)
    finally:
        ops.FORCE_EAGER = old
    assert v.numel() == k and i.unique().numel() == k
    thresh = x.abs().topk(k)[0][-1]
    assert (v.abs() >= thresh - 1e-6).all()
    assert torch.equal(v, x[i.long()])
