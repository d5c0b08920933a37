# -*- coding: utf-8 -*-
"""HIP kernel pack numerics: every CDNA4 kernel vs the eager torch fp32
reference (the same eager code path the CPU tests pin against the reference
formulas)."""
import pytest
import torch

import fedtorch_amd.ops as ops

pytestmark = pytest.mark.gpu

N = 64 * 1024 + 64  # arena-aligned, > one wave-block


def _pair(n=N, seed=0):
    torch.manual_seed(seed)
    return (torch.randn(n, device='cuda'), torch.randn(n, device='cuda'))


def eager(fn, *t, **kw):
    """run the op's eager path on CPU clones."""
    cpu = [x.cpu() if isinstance(x, torch.Tensor) else x for x in t]
    old = ops.FORCE_EAGER
    ops.FORCE_EAGER = True
    try:
        out = fn(*cpu, **kw)
    finally:
        ops.FORCE_EAGER = old
    return out, cpu


def test_ext_is_built():
    assert ops.hip_available(), 'HIP kernel pack must be built on GPU boxes'


def test_weighted_diff_restore_gpu():
    s, c = _pair()
    out = torch.zeros_like(s)
    c0 = c.clone()
    ops.weighted_diff_restore(s, c, out, 0.3)
    assert torch.allclose(out.cpu(), ((s - c0) * 0.3).cpu(), atol=1e-6)
    assert torch.equal(c, s)


def test_fused_sgd_gpu_matches_eager():
    torch.manual_seed(1)
    n = N
    p = torch.randn(n, device='cuda')
    g = torch.randn(n, device='cuda')
    in_buf = torch.zeros_like(p)
    server = torch.randn(n, device='cuda')
    pc, gc, bc, sc = p.cpu(), g.cpu(), in_buf.cpu(), server.cpu()
    kw = dict(lr=0.1, scale=1.0, weight_decay=0.01, in_momentum=0.9,
              out_momentum=0.0, dampening=0.0, nesterov=True, apply_lr=True,
              apply_in_momentum=True, apply_out_momentum=False,
              prox_mu=0.05, wd_numel=n - 256)
    # two steps to exercise first_in then steady state
    ops.fused_sgd_step(p, g, in_buf=in_buf, server=server, first_in=True,
                       **kw)
    ops.fused_sgd_step(p, g, in_buf=in_buf, server=server, first_in=False,
                       **kw)
    old = ops.FORCE_EAGER
    ops.FORCE_EAGER = True
    try:
        ops.fused_sgd_step(pc, gc, in_buf=bc, server=sc, first_in=True, **kw)
        ops.fused_sgd_step(pc, gc, in_buf=bc, server=sc, first_in=False, **kw)
    finally:
        ops.FORCE_EAGER = old
    assert torch.allclose(p.cpu(), pc, atol=1e-5)
    assert torch.allclose(in_buf.cpu(), bc, atol=1e-5)


def test_fused_sgd_gpu_corrections():
    torch.manual_seed(2)
    n = 4096
    p = torch.randn(n, device='cuda')
    g = torch.randn(n, device='cuda')
    delta = torch.randn(n, device='cuda')
    cs = torch.randn(n, device='cuda')
    cc = torch.randn(n, device='cuda')
    p0 = p.clone()
    ops.fused_sgd_step(p, g, delta=delta, ctrl_server=cs, ctrl_client=cc,
                       lr=0.2, scale=1.0, weight_decay=0.0, in_momentum=0.0,
                       out_momentum=0.0, dampening=0.0, nesterov=False,
                       apply_lr=True, apply_in_momentum=False,
                       apply_out_momentum=False)
    expected = p0 - 0.2 * (g - delta + cs - cc)
    assert torch.allclose(p.cpu(), expected.cpu(), atol=1e-5)


@pytest.mark.parametrize('bits', [8, 16])
def test_quantize_gpu_matches_eager(bits):
    x = _pair(seed=3)[0]
    q, info = ops.quantize(x, bits)
    (qe, infoe), _ = eager(ops.quantize, x, num_bits=bits)
    assert torch.allclose(info.cpu(), infoe, rtol=1e-5, atol=1e-5)
    # mean differs in last ulp between GPU tree-sum and CPU serial sum;
    # codes may differ by 1 on exact rounding boundaries — bound the count.
    diff = (q.cpu().long() - qe.long()).abs()
    assert (diff > 1).sum().item() == 0
    assert (diff == 1).float().mean().item() < 1e-3
    # round-trip bound: scale/2 quantization + up to 1 grid unit of clipping
    # from the reference's truncated zero-point (`flow_utils.py:183-192`)
    xr = ops.dequantize(q, info)
    assert (x - xr).abs().max().item() <= info[0].item() * 1.5 + 1e-6


def test_dequant_accumulate_gpu():
    torch.manual_seed(4)
    xs = torch.randn(4, 8192, device='cuda')
    qs, infos = [], []
    for k in range(4):
        q, i = ops.quantize(xs[k], 8)
        qs.append(q)
        infos.append(i)
    qs = torch.stack(qs)
    infos = torch.stack(infos)
    out = torch.zeros(8192, device='cuda')
    ops.dequant_accumulate(qs, infos, out)
    expected = sum(ops.dequantize(qs[k], infos[k]) for k in range(4))
    assert torch.allclose(out, expected, atol=1e-5)


def test_topk_gpu_exact_selection():
    torch.manual_seed(5)
    x = torch.randn(200000, device='cuda')
    k = 5000
    v, i = ops.topk_compress(x, k)
    vr, ir = x.abs().topk(k)
    assert v.shape[0] == k
    # selected |values| must match torch.topk's (sets equal up to threshold
    # ties)
    assert torch.allclose(v.abs().sort(descending=True)[0], vr,
                          atol=1e-6)
    # indices consistent with values
    assert torch.equal(v, x[i.long()])
    # no duplicate indices
    assert i.unique().numel() == k


def test_topk_gpu_ties():
    x = torch.ones(4096, device='cuda')
    v, i = ops.topk_compress(x, 100)
    assert v.shape[0] == 100
    assert i.unique().numel() == 100
    assert torch.allclose(v, torch.ones(100, device='cuda'))


def test_scatter_accumulate_gpu():
    torch.manual_seed(6)
    n, k = 65536, 2048
    vs, idxs = [], []
    for s in range(5):
        x = torch.randn(n, device='cuda')
        v, i = ops.topk_compress(x, k)
        vs.append(v)
        idxs.append(i)
    vs = torch.stack(vs)
    idxs = torch.stack(idxs)
    out = torch.empty(n, device='cuda')
    ops.scatter_accumulate(out, vs, idxs)
    expected = torch.zeros(n, device='cuda')
    for s in range(5):
        expected.scatter_add_(0, idxs[s].long(), vs[s])
    assert torch.allclose(out, expected, atol=1e-5)


def test_elementwise_gpu():
    a, b = _pair(4096, seed=7)
    y = a.clone()
    ops.axpby(y, b, a=0.3, b=0.7)
    assert torch.allclose(y, 0.7 * a + 0.3 * b, atol=1e-6)

    mem, g = _pair(4096, seed=8)
    mem0 = mem.clone()
    d = torch.randn(4096, device='cuda')
    ops.error_feedback_update(mem, g, d, 2.0)
    assert torch.allclose(mem, mem0 + 2 * g - d, atol=1e-5)

    out = torch.zeros(4096, device='cuda')
    ops.blend(out, a, b, 0.25)
    assert torch.allclose(out, 0.25 * a + 0.75 * b, atol=1e-6)

    lf, pf = _pair(8192, seed=9)
    lg, pg = _pair(8192, seed=10)
    ga = ops.alpha_grad(lf, pf, lg, pg, 0.4)
    expected = torch.dot(pf - lf, 0.4 * pg + 0.6 * lg).item() + 0.02 * 0.4
    assert abs(ga - expected) < abs(expected) * 1e-3 + 1e-3


def test_scaffold_delta_gpu():
    cc, cs = _pair(4096, seed=11)
    s, c = _pair(4096, seed=12)
    out = torch.zeros(4096, device='cuda')
    ops.scaffold_control_update(out, cc, cs, s, c, 1.5)
    assert torch.allclose(out, cc - cs + 1.5 * (s - c), atol=1e-5)

    delta = torch.randn(4096, device='cuda')
    d0 = delta.clone()
    agg = torch.randn(4096, device='cuda')
    ops.delta_update(delta, s, agg, c, 0.25)
    assert torch.allclose(delta, d0 + 0.25 * (s - agg - c), atol=1e-5)


def test_multi_diff_accumulate_gpu():
    torch.manual_seed(13)
    C, n = 7, 8192
    server = torch.randn(n, device='cuda')
    replicas = torch.randn(C, n, device='cuda')
    w = torch.rand(C, device='cuda')
    w[2] = 0.0
    out = torch.empty(n, device='cuda')
    ops.multi_diff_accumulate(server, replicas, w, out)
    expected = ((server.unsqueeze(0) - replicas) * w.view(-1, 1)).sum(0)
    assert torch.allclose(out, expected, atol=1e-4)


def test_fused_batchnorm_matches_torch():
    """fused BN fwd/bwd vs nn.BatchNorm2d on identical fp32 inputs."""
    import torch.nn as nn
    from fedtorch_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(20)
    N, C, H, W = 16, 32, 14, 14
    x = torch.randn(N, C, H, W, device='cuda')
    ref_bn = nn.BatchNorm2d(C).cuda()
    fus_bn = FusedBatchNorm2d(C).cuda()
    fus_bn.load_state_dict(ref_bn.state_dict())
    ref_bn.weight.data.uniform_(0.5, 1.5)
    ref_bn.bias.data.uniform_(-0.5, 0.5)
    fus_bn.weight.data.copy_(ref_bn.weight.data)
    fus_bn.bias.data.copy_(ref_bn.bias.data)

    xr = x.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    yr = ref_bn(xr)
    yf = fus_bn(xf)
    assert torch.allclose(yf, yr, atol=2e-5), \
        (yf - yr).abs().max().item()
    g = torch.randn_like(yr)
    yr.backward(g)
    yf.backward(g)
    assert torch.allclose(xf.grad, xr.grad, atol=2e-4)
    assert torch.allclose(fus_bn.weight.grad, ref_bn.weight.grad, atol=1e-3)
    assert torch.allclose(fus_bn.bias.grad, ref_bn.bias.grad, atol=1e-3)
    assert torch.allclose(fus_bn.running_mean, ref_bn.running_mean,
                          atol=1e-5)
    assert torch.allclose(fus_bn.running_var, ref_bn.running_var, atol=1e-4)


def test_fused_batchnorm_bf16_io():
    from fedtorch_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(21)
    x = torch.randn(8, 16, 8, 8, device='cuda', dtype=torch.bfloat16)
    bn = FusedBatchNorm2d(16).cuda()
    xr = x.clone().requires_grad_(True)
    y = bn(xr)
    assert y.dtype == torch.bfloat16
    y.float().square().mean().backward()
    assert xr.grad is not None and torch.isfinite(xr.grad.float()).all()
    # fp32 reference on the same values
    import torch.nn as nn
    ref = nn.BatchNorm2d(16).cuda()
    yr = ref(x.float())
    assert torch.allclose(y.float(), yr, atol=0.05)


def test_fused_bn_relu_matches_torch():
    import torch.nn as nn
    from fedtorch_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(22)
    N, C, H, W = 16, 32, 8, 8
    x = torch.randn(N, C, H, W, device='cuda')
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    fus = FusedBatchNorm2d(C).cuda()
    fus.fuse_relu = True
    fus.load_state_dict(ref.state_dict())
    xr = x.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    yr = torch.relu(ref(xr))
    yf = fus(xf)
    assert torch.allclose(yf, yr, atol=2e-5)
    g = torch.randn_like(yr)
    yr.backward(g)
    yf.backward(g)
    assert torch.allclose(xf.grad, xr.grad, atol=2e-4)
    assert torch.allclose(fus.weight.grad, ref.weight.grad, atol=1e-3)
    assert torch.allclose(fus.bias.grad, ref.bias.grad, atol=1e-3)


@pytest.mark.parametrize('hw', [8, 7])  # 8x8 → vectorized, 7x7 → scalar path
def test_fused_bn_add_relu_matches_torch(hw):
    """BNAddReLU: relu(bn(x) + res) fused fwd + dres bwd vs eager."""
    import torch.nn as nn
    from fedtorch_amd.ops.batchnorm import BNAddReLU, convert_to_fused_bn
    torch.manual_seed(23)
    N, C = 16, 32
    x = torch.randn(N, C, hw, hw, device='cuda')
    res = torch.randn(N, C, hw, hw, device='cuda')
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    fus = BNAddReLU(C).cuda()
    fus.bn.load_state_dict(ref.state_dict())
    convert_to_fused_bn(fus)
    xr = x.clone().requires_grad_(True)
    rr = res.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    rf = res.clone().requires_grad_(True)
    yr = torch.relu(ref(xr) + rr)
    yf = fus(xf, rf)
    assert torch.allclose(yf, yr, atol=2e-5)
    g = torch.randn_like(yr)
    yr.backward(g)
    yf.backward(g)
    assert torch.allclose(xf.grad, xr.grad, atol=2e-4)
    assert torch.allclose(rf.grad, rr.grad, atol=2e-5)
    assert torch.allclose(fus.bn.weight.grad, ref.weight.grad, atol=1e-3)
    assert torch.allclose(fus.bn.bias.grad, ref.bias.grad, atol=1e-3)
    assert torch.allclose(fus.bn.running_mean, ref.running_mean, atol=1e-5)


def test_gather_grads_matches_attached():
    """Stolen-grad gather (graph path) fills the grad arena with exactly
    what attached-view accumulation produces."""
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    torch.manual_seed(24)

    def make():
        torch.manual_seed(7)
        return nn.Sequential(nn.Linear(33, 65), nn.ReLU(),
                             nn.Linear(65, 10)).cuda()

    x = torch.randn(16, 33, device='cuda')
    y = torch.randint(0, 10, (16,), device='cuda')
    crit = nn.CrossEntropyLoss()

    m1 = make()
    a1 = Arena(m1)
    a1.zero_grad()
    crit(m1(x), y).backward()
    ref = a1.grad.clone()

    m2 = make()
    a2 = Arena(m2)
    a2.detach_grads()
    crit(m2(x), y).backward()
    a2.gather_grads()
    torch.cuda.synchronize()
    assert torch.allclose(a2.grad, ref, atol=1e-6)
    # second call reuses the cached chunk table
    assert a2._gather_state
    a2.gather_grads()
    torch.cuda.synchronize()
    assert torch.allclose(a2.grad, ref, atol=1e-6)


@pytest.mark.parametrize('C', [16, 32, 64])
@pytest.mark.parametrize('mode', ['plain', 'relu', 'addrelu'])
def test_fused_bn_nhwc_matches_torch(C, mode):
    """NHWC (channels_last) BN kernels vs eager reference."""
    import torch.nn as nn
    from fedtorch_amd.ops.batchnorm import FusedBatchNorm2d, BNAddReLU, \
        convert_to_fused_bn
    torch.manual_seed(31)
    N, H, W = 8, 10, 10
    cl = torch.channels_last
    x = torch.randn(N, C, H, W, device='cuda').contiguous(memory_format=cl)
    ref = nn.BatchNorm2d(C).cuda()
    ref.weight.data.uniform_(0.5, 1.5)
    ref.bias.data.uniform_(-0.5, 0.5)
    xr = x.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    if mode == 'addrelu':
        res = torch.randn_like(x).contiguous(memory_format=cl)
        rr = res.clone().requires_grad_(True)
        rf = res.clone().requires_grad_(True)
        fus = BNAddReLU(C).cuda()
        fus.bn.load_state_dict(ref.state_dict())
        convert_to_fused_bn(fus)
        yr = torch.relu(ref(xr) + rr)
        yf = fus(xf, rf)
        fw, fb = fus.bn.weight, fus.bn.bias
        frm, frv = fus.bn.running_mean, fus.bn.running_var
    else:
        fus = FusedBatchNorm2d(C).cuda()
        fus.fuse_relu = mode == 'relu'
        fus.load_state_dict(ref.state_dict())
        yr = ref(xr)
        if mode == 'relu':
            yr = torch.relu(yr)
        yf = fus(xf)
        fw, fb = fus.weight, fus.bias
        frm, frv = fus.running_mean, fus.running_var
    assert yf.is_contiguous(memory_format=cl)
    assert torch.allclose(yf, yr, atol=3e-5), (yf - yr).abs().max().item()
    g = torch.randn_like(yr)
    yr.backward(g)
    yf.backward(g)
    assert torch.allclose(xf.grad, xr.grad, atol=3e-4)
    if mode == 'addrelu':
        assert torch.allclose(rf.grad, rr.grad, atol=3e-5)
    assert torch.allclose(fw.grad, ref.weight.grad, atol=2e-3)
    assert torch.allclose(fb.grad, ref.bias.grad, atol=2e-3)
    assert torch.allclose(frm, ref.running_mean, atol=1e-5)
    assert torch.allclose(frv, ref.running_var, atol=1e-4)


@pytest.mark.parametrize('xdtype', ['float32', 'bfloat16'])
def test_nhwc_stem_conv_matches_torch(xdtype):
    """Custom NHWC stem conv (fwd + wrw) vs F.conv2d."""
    import torch.nn.functional as F
    dt = getattr(torch, xdtype)
    torch.manual_seed(32)
    cl = torch.channels_last
    N, Ci, H, W, Co = 8, 3, 32, 32, 16
    x = torch.randn(N, Ci, H, W, device='cuda', dtype=dt).contiguous(
        memory_format=cl)
    w = torch.randn(Co, Ci, 3, 3, device='cuda').contiguous(
        memory_format=cl) * 0.1
    y = ops._C.stem_conv_fwd(x, w, False)
    ref = F.conv2d(x.float(), w, padding=1)
    tol = 1e-4 if xdtype == 'float32' else 0.05
    assert y.is_contiguous(memory_format=cl)
    assert torch.allclose(y, ref, atol=tol), (y - ref).abs().max().item()

    dy = torch.randn(N, Co, H, W, device='cuda', dtype=dt).contiguous(
        memory_format=cl)
    dw = ops._C.stem_conv_wrw(dy, x)
    wr = w.clone().requires_grad_(True)
    F.conv2d(x.float(), wr, padding=1).backward(dy.float())
    wtol = 2e-3 if xdtype == 'float32' else 2.0
    assert torch.allclose(dw, wr.grad, atol=wtol, rtol=1e-2), \
        (dw - wr.grad).abs().max().item()


def test_nhwc_resnet20_step_matches_nchw():
    """Full resnet20: one bf16 fwd/bwd in channels_last (fused NHWC BN +
    custom stem) tracks the NCHW eager path."""
    from types import SimpleNamespace
    from fedtorch_amd.components.models.resnet import resnet
    from fedtorch_amd.ops.batchnorm import convert_to_fused_bn
    from fedtorch_amd.ops.stemconv import convert_stem, NhwcStemConv
    from fedtorch_amd.parallel.arena import Arena
    a = SimpleNamespace(arch='resnet20', data='cifar10')
    torch.manual_seed(33)
    x = torch.randn(16, 3, 32, 32, device='cuda')
    y = torch.randint(0, 10, (16,), device='cuda')
    crit = torch.nn.CrossEntropyLoss()

    def build(cl):
        torch.manual_seed(5)
        m = resnet(a).cuda()
        if cl:
            m = m.to(memory_format=torch.channels_last)
            convert_to_fused_bn(m)
            convert_stem(m)
        return m, Arena(m)

    m_ref, a_ref = build(False)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        loss_ref = crit(m_ref(x), y)
    loss_ref.backward()

    m_cl, a_cl = build(True)
    assert isinstance(m_cl.conv1, NhwcStemConv)
    xc = x.contiguous(memory_format=torch.channels_last)
    with torch.autocast('cuda', dtype=torch.bfloat16):
        loss_cl = crit(m_cl(xc), y)
    loss_cl.backward()
    assert abs(loss_cl.item() - loss_ref.item()) < 0.05
    # grad arenas carry the same energy (bf16 noise aside); element order
    # differs (NHWC packing) so compare norms
    n_ref = a_ref.grad.norm().item()
    n_cl = a_cl.grad.norm().item()
    assert abs(n_cl - n_ref) / max(n_ref, 1e-6) < 0.1, (n_ref, n_cl)


def test_bf16_compute_arena_step_matches_fp32():
    """bf16 compute twin (fp32 master): a fwd/bwd/step through bf16 views +
    gather tracks the fp32+autocast path, and the twin stays in sync."""
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    from fedtorch_amd.components.optim.sgd import FusedSGD

    def build():
        torch.manual_seed(9)
        return nn.Sequential(nn.Linear(32, 64), nn.ReLU(),
                             nn.Linear(64, 10)).cuda()

    x = torch.randn(64, 32, device='cuda')
    y = torch.randint(0, 10, (64,), device='cuda')
    crit = nn.CrossEntropyLoss()

    m1 = build()
    a1 = Arena(m1)
    o1 = FusedSGD(a1, lr=0.1)
    a1.zero_grad()
    with torch.autocast('cuda', dtype=torch.bfloat16):
        crit(m1(x), y).backward()
    o1.step(apply_lr=True)

    m2 = build()
    a2 = Arena(m2)
    o2 = FusedSGD(a2, lr=0.1)
    a2.enable_bf16_compute()
    assert m2[0].weight.dtype == torch.bfloat16
    a2.detach_grads()
    with torch.autocast('cuda', dtype=torch.bfloat16):
        crit(m2(x), y).backward()
    a2.gather_grads()
    o2.step(apply_lr=True)
    torch.cuda.synchronize()
    # masters agree to bf16-grad noise
    assert torch.allclose(a2.flat, a1.flat, atol=2e-3), \
        (a2.flat - a1.flat).abs().max().item()
    # twin refreshed by the fused step in the same pass
    assert torch.allclose(a2.half_flat.float(), a2.flat, atol=1e-2)
    # direct master mutation + sync_half
    a2.flat.mul_(0.5)
    a2.sync_half()
    torch.cuda.synchronize()
    assert torch.allclose(a2.half_flat.float(), a2.flat, atol=1e-2)


def test_fused_bn_nbt_lazy_flush():
    """num_batches_tracked counts on the host and flushes into the buffer
    when the state_dict is read (no per-step GPU kernel)."""
    from fedtorch_amd.ops.batchnorm import FusedBatchNorm2d
    bn = FusedBatchNorm2d(8).cuda()
    x = torch.randn(4, 8, 8, 8, device='cuda')
    for _ in range(3):
        bn(x)
    sd = bn.state_dict()
    assert int(sd['num_batches_tracked']) == 3
    assert bn._nbt_pending == 0


def test_mfma_tile_gemm():
    """mfma_f32_16x16x32_bf16 fragment-layout probe: D = A @ B with the
    assumed lane mappings must equal torch matmul (asymmetric operands)."""
    torch.manual_seed(40)
    A = (torch.randn(16, 32) * 0.5).bfloat16().cuda()
    B = (torch.arange(32 * 16).reshape(32, 16).float() / 256.0
         + torch.randn(32, 16)).bfloat16().cuda()
    D = ops._C.mfma_probe(A.contiguous(), B.contiguous())
    ref = A.float() @ B.float()
    assert torch.allclose(D, ref, atol=1e-2, rtol=1e-2), \
        (D - ref).abs().max().item()


def test_bench_contract():
    """bench.py emits the driver-contract JSON line and runs the native
    path (subprocess, tiny step count)."""
    import json
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, 'bench.py'), '--steps', '12',
         '--warmup', '4', '--batch', '64'],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith('{')][-1]
    d = json.loads(line)
    assert d['metric'] == 'samples/sec'
    assert d['value'] > 0
    assert d['n_gpus'] == 1
    assert d['dtype'] == 'bf16'
    assert d['data'] == 'synthetic'
    assert d['config']['model'] == 'resnet20'


@pytest.mark.parametrize('C,W', [(16, 32), (32, 16), (64, 8)])
def test_conv3x3_wrw_matches_torch(C, W):
    """MFMA wrw kernel vs F.conv2d weight gradient (bf16, channels_last)."""
    import torch.nn.functional as F
    torch.manual_seed(41)
    cl = torch.channels_last
    N, H = 8, W
    x = (torch.randn(N, C, H, W, device='cuda') * 0.5).bfloat16() \
        .contiguous(memory_format=cl)
    dy = (torch.randn(N, C, H, W, device='cuda') * 0.5).bfloat16() \
        .contiguous(memory_format=cl)
    dw = ops._C.conv3x3_wrw(dy, x)
    assert dw.dtype == torch.bfloat16
    assert dw.is_contiguous(memory_format=cl)
    w = torch.zeros(C, C, 3, 3, device='cuda', requires_grad=True)
    F.conv2d(x.float(), w, padding=1).backward(dy.float())
    ref = w.grad
    err = (dw.float() - ref).abs()
    tol = ref.abs().max().item() * 0.02 + 0.5
    assert err.max().item() < tol, (err.max().item(), ref.abs().max().item())


def test_nhwc_conv3x3_module_grads():
    """NhwcConv3x3 end-to-end: dx and dw match the stock conv autograd."""
    import torch.nn.functional as F
    import fedtorch_amd.ops.conv3x3 as c3
    from fedtorch_amd.ops.conv3x3 import NhwcConv3x3
    c3._ENABLED = True  # custom wrw path is env-gated off by default
    torch.manual_seed(42)
    cl = torch.channels_last
    N, C, H, W = 8, 16, 32, 32
    conv = NhwcConv3x3(C, C, 3, padding=1, bias=False).cuda() \
        .to(memory_format=cl)
    conv.weight.data = conv.weight.data.bfloat16()
    x = torch.randn(N, C, H, W, device='cuda').bfloat16() \
        .contiguous(memory_format=cl).requires_grad_(True)
    y = conv(x)
    g = torch.randn_like(y)
    y.backward(g)
    # reference via stock conv on the same values
    xr = x.detach().clone().requires_grad_(True)
    wr = conv.weight.detach().clone().requires_grad_(True)
    F.conv2d(xr, wr, padding=1).backward(g)
    assert torch.allclose(x.grad.float(), xr.grad.float(), atol=1e-2,
                          rtol=1e-2)
    ref = wr.grad.float()
    tol = ref.abs().max().item() * 0.02 + 0.5
    assert (conv.weight.grad.float() - ref).abs().max().item() < tol


def test_fedadam_normalize_gpu_matches_eager():
    """fedadam_norm_kernel: segmented per-tensor norm + v update + scale in
    one launch vs the eager per-segment formula."""
    torch.manual_seed(5)
    # uneven segments incl. a tiny one and a multi-block one
    sizes = [7, 300, 40000, 1024]
    offs, cur = [], 0
    for sz in sizes:
        offs.append((cur, cur + sz))
        cur += sz + 13  # pad gaps like the arena
    g = torch.randn(cur, device='cuda')
    seg = torch.tensor(offs, dtype=torch.long, device='cuda')
    v = torch.full((len(sizes),), 0.25, device='cuda')
    g_ref, v_ref = g.cpu().clone(), v.cpu().clone()
    ops.fedadam_normalize(g, seg, v, 0.9, 0.1)
    ops.FORCE_EAGER = True
    try:
        ops.fedadam_normalize(g_ref, seg.cpu(), v_ref, 0.9, 0.1)
    finally:
        ops.FORCE_EAGER = False
    assert torch.allclose(v.cpu(), v_ref, atol=1e-5, rtol=1e-5)
    assert torch.allclose(g.cpu(), g_ref, atol=1e-5, rtol=1e-5)


def test_conv3x3_bn_fused_fwd_matches_reference():
    """conv3x3_bn_fwd + bn_fwd_train_part (stats from the conv epilogue)
    must match the fp32 conv + BN reference on the same bf16 inputs."""
    import torch.nn.functional as F
    CL = torch.channels_last
    torch.manual_seed(3)
    for C, H in ((16, 32), (32, 16), (64, 8)):
        N, W = 64, H
        x = torch.randn(N, C, H, W, device='cuda').to(
            memory_format=CL).bfloat16()
        w = (torch.randn(C, C, 3, 3, device='cuda') / (3 * C) ** 0.5).to(
            memory_format=CL).bfloat16()
        gamma = torch.rand(C, device='cuda') + 0.5
        beta = torch.randn(C, device='cuda') * 0.1
        grid = N * (H // 8) * (2 if C == 64 else 1)
        part = torch.empty(grid, C, 2, device='cuda')
        e = torch.empty(0, device='cuda')
        y = ops._C.conv3x3_bn_fwd(x, w, part, e, e, e, False)
        z, sm, siv, _ = ops._C.bn_fwd_train_part(
            y, part, gamma, beta, e, e, 1e-5, 0.1, True, e)
        ref_y = F.conv2d(x.float(), w.float(), None, 1, 1)
        ref = F.batch_norm(ref_y, None, None, gamma, beta, True, 0.1, 1e-5)
        ref = torch.relu(ref)
        err = (z.float() - ref).abs().max().item()
        assert err < 0.12, 'C%d: fused conv+BN err %.4f' % (C, err)
        m_ref = ref_y.mean(dim=(0, 2, 3))
        assert torch.allclose(sm, m_ref, atol=2e-3), 'mean mismatch'


def test_nhwc_conv_module_fused_path_autograd():
    """NhwcConv3x3 -> BNReLU with the fused fwd + part handoff: full
    fwd/bwd vs the eager fp32 module chain (same bf16 inputs)."""
    import torch.nn.functional as F
    from fedtorch_amd.ops.conv3x3 import NhwcConv3x3
    from fedtorch_amd.ops.batchnorm import BNReLU, convert_to_fused_bn
    CL = torch.channels_last
    torch.manual_seed(4)
    C, H = 16, 32
    conv = NhwcConv3x3(C, C, kernel_size=3, stride=1, padding=1,
                       bias=False).cuda().to(memory_format=CL)
    bn = convert_to_fused_bn(torch.nn.Sequential(BNReLU(C))).cuda()
    conv_w32 = conv.weight.detach().float().clone()
    conv.weight.data = conv.weight.data.bfloat16()
    conv.train(); bn.train()
    x = torch.randn(64, C, H, H, device='cuda').to(
        memory_format=CL).bfloat16().requires_grad_(True)
    z = bn(conv(x))
    loss = (z.float() ** 2).mean()
    loss.backward()
    # reference in fp32 on the same bf16 values
    xr = x.detach().float().requires_grad_(True)
    bnr = torch.nn.BatchNorm2d(C).cuda()
    with torch.no_grad():
        bnr.weight.copy_(bn[0].bn.weight)
        bnr.bias.copy_(bn[0].bn.bias)
    yr = F.conv2d(xr, conv_w32, None, 1, 1)
    zr = torch.relu(bnr(yr))
    lr_ = (zr ** 2).mean()
    lr_.backward()
    assert (z.float() - zr).abs().max().item() < 0.05
    assert (x.grad.float() - xr.grad).abs().max().item() < 0.02


def test_conv3x3_wrw2_numerics():
    """wrw v2 (transposed-LDS staging) vs the fp32 aten reference."""
    CL = torch.channels_last
    torch.manual_seed(9)
    for C, W in ((16, 32), (32, 16), (64, 8)):
        x = torch.randn(64, C, W, W, device='cuda').bfloat16().contiguous(
            memory_format=CL)
        dy = torch.randn(64, C, W, W, device='cuda').bfloat16().contiguous(
            memory_format=CL)
        w = torch.randn(C, C, 3, 3, device='cuda').bfloat16().contiguous(
            memory_format=CL)
        dw = ops._C.conv3x3_wrw2(dy, x)
        ref = torch.ops.aten.convolution_backward(
            dy.float(), x.float(), w.float(), None, [1, 1], [1, 1], [1, 1],
            False, [0, 0], 1, [False, True, False])[1]
        rel = (dw.float() - ref).abs().max().item() / ref.abs().max().item()
        assert rel < 0.01, 'C%d rel %.5f' % (C, rel)


def test_conv3x3_dgrad_numerics():
    """custom backward-data (fwd kernel's mirror) vs fp32 aten."""
    CL = torch.channels_last
    torch.manual_seed(13)
    for C, W in ((16, 32), (32, 16), (64, 8)):
        dy = torch.randn(64, C, W, W, device='cuda').bfloat16().contiguous(
            memory_format=CL)
        x = torch.randn(64, C, W, W, device='cuda').bfloat16().contiguous(
            memory_format=CL)
        w = (torch.randn(C, C, 3, 3, device='cuda') / C).bfloat16()
        w = w.contiguous(memory_format=CL)
        dx = ops._C.conv3x3_dgrad(dy, w)
        ref = torch.ops.aten.convolution_backward(
            dy.float(), x.float(), w.float(), None, [1, 1], [1, 1], [1, 1],
            False, [0, 0], 1, [True, False, False])[0]
        rel = (dx.float() - ref).abs().max().item() / ref.abs().max().item()
        assert rel < 0.01, 'C%d rel %.5f' % (C, rel)


def test_bn_defer_backward_matches_eager():
    """Deferred BN backward (dx applied in the conv dgrad staging) must
    match the eager bn_bwd + conv backward chain on a conv->BNReLU->conv
    stack, including weight/bias grads."""
    import os
    import torch.nn as nn
    from fedtorch_amd.ops.conv3x3 import NhwcConv3x3
    from fedtorch_amd.ops import batchnorm as bnm
    from fedtorch_amd.ops import conv3x3 as c3
    CL = torch.channels_last

    def build():
        torch.manual_seed(21)
        m = nn.Sequential(
            NhwcConv3x3(16, 16, 3, padding=1, bias=False),
            bnm.BNReLU(16),
            NhwcConv3x3(16, 16, 3, padding=1, bias=False),
        ).cuda().to(memory_format=CL)
        bnm.convert_to_fused_bn(m)
        for p_ in m.parameters():
            if p_.dim() == 4:
                p_.data = p_.data.bfloat16()
        m.train()
        return m

    def run(defer):
        old = c3._BNDEFER_ENABLED
        c3._BNDEFER_ENABLED = defer
        try:
            m = build()
            torch.manual_seed(33)
            x = torch.randn(64, 16, 32, 32, device='cuda').to(
                memory_format=CL).bfloat16().requires_grad_(True)
            out = m(x)
            loss = (out.float() ** 2).mean()
            loss.backward()
            grads = [x.grad.float().clone()] + \
                [p_.grad.float().clone() for p_ in m.parameters()]
        finally:
            c3._BNDEFER_ENABLED = old
        return grads

    g_def = run(True)
    g_eag = run(False)
    assert not c3._BNBWD_TAGS, 'side table must drain'
    for i, (a, b) in enumerate(zip(g_def, g_eag)):
        scale = b.abs().max().item() + 1e-6
        rel = (a - b).abs().max().item() / scale
        assert rel < 0.02, 'grad %d rel %.5f' % (i, rel)


def test_conv3x3_s2_fused_fwd():
    """stride-2 transition conv (fused BN-stats epilogue) vs fp32 ref."""
    import torch.nn.functional as F
    CL = torch.channels_last
    torch.manual_seed(15)
    for Ci, Wi in ((16, 32), (32, 16)):
        Co, N = 2 * Ci, 64
        x = torch.randn(N, Ci, Wi, Wi, device='cuda').to(
            memory_format=CL).bfloat16()
        w = (torch.randn(Co, Ci, 3, 3, device='cuda') / (3 * Ci) ** .5).to(
            memory_format=CL).bfloat16()
        grid = N * (Wi // 2 // 8) * (Co // 32)
        part = torch.zeros(grid, Co, 2, device='cuda')
        y = ops._C.conv3x3s2_bn_fwd(x, w, part)
        ref = F.conv2d(x.float(), w.float(), None, 2, 1)
        rel = (y.float() - ref).abs().max().item() / ref.abs().max().item()
        assert rel < 0.01, 'Ci%d rel %.5f' % (Ci, rel)
        s_ref = y.float().sum(dim=(0, 2, 3))
        e = (part[:, :, 0].sum(0) - s_ref).abs().max().item() / (
            (y.float() ** 2).sum().item() ** 0.5 + 1e-6)
        assert e < 1e-2, 'stats %.5f' % e


def test_bn_defer_full_resnet_matches_eager():
    """Deferred BN backward across a FULL ResNet-20 step (incl. stride-2
    transitions, residual forks and BNAddReLU dres) vs the eager path."""
    from types import SimpleNamespace
    from fedtorch_amd.ops import conv3x3 as c3
    from fedtorch_amd.components.models.resnet import resnet
    from fedtorch_amd.ops.batchnorm import convert_to_fused_bn
    CL = torch.channels_last

    def run(defer):
        old = c3._BNDEFER_ENABLED
        c3._BNDEFER_ENABLED = defer
        try:
            torch.manual_seed(41)
            args = SimpleNamespace(arch='resnet20', data='cifar10')
            m = resnet(args).cuda().to(memory_format=CL)
            convert_to_fused_bn(m)
            for p_ in m.parameters():
                if p_.dim() == 4:
                    p_.data = p_.data.bfloat16()
            m.train()
            torch.manual_seed(42)
            x = torch.randn(32, 3, 32, 32, device='cuda').to(
                memory_format=CL)
            with torch.autocast('cuda', dtype=torch.bfloat16):
                out = m(x)
                loss = out.float().square().mean()
            loss.backward()
            gs = [p_.grad.float().clone() for p_ in m.parameters()
                  if p_.grad is not None]
        finally:
            c3._BNDEFER_ENABLED = old
        return gs

    g_d = run(True)
    g_e = run(False)
    assert not c3._BNBWD_TAGS, 'side table must drain'
    assert len(g_d) == len(g_e)
    for i, (a, b) in enumerate(zip(g_d, g_e)):
        scale = b.abs().max().item() + 1e-5
        rel = (a - b).abs().max().item() / scale
        assert rel < 0.05, 'param %d rel %.4f' % (i, rel)


def test_conv1x1_s2_gemm_path():
    """1x1/stride-2 downsample on the GEMM path: fwd + both grads vs
    the fp32 conv reference."""
    import torch.nn.functional as F
    from fedtorch_amd.ops import conv3x3 as c3
    from fedtorch_amd.ops.conv3x3 import NhwcConv1x1S2
    CL = torch.channels_last
    old = c3._CONV1X1_GEMM
    c3._CONV1X1_GEMM = True
    torch.manual_seed(17)
    for Ci, Wi in ((16, 32), (32, 16)):
        Co = 2 * Ci
        m = NhwcConv1x1S2(Ci, Co, kernel_size=1, stride=2,
                          bias=False).cuda().to(memory_format=CL)
        m.weight.data = m.weight.data.bfloat16()
        m.train()
        x = torch.randn(64, Ci, Wi, Wi, device='cuda').to(
            memory_format=CL).bfloat16().requires_grad_(True)
        y = m(x)
        loss = (y.float() ** 2).mean()
        loss.backward()
        xr = x.detach().float().requires_grad_(True)
        w32 = m.weight.detach().float().requires_grad_(True)
        yr = F.conv2d(xr, w32, None, 2, 0)
        (yr ** 2).mean().backward()
        assert (y.float() - yr).abs().max().item() < 0.05
        assert (x.grad.float() - xr.grad).abs().max().item() < 0.02
        rel = (m.weight.grad.float() - w32.grad).abs().max().item() / \
            w32.grad.abs().max().item()
        assert rel < 0.02, 'dw rel %.5f' % rel
    c3._CONV1X1_GEMM = old
