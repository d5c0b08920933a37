# -*- coding: utf-8 -*-
"""Arena structure + fused SGD vs a literal reference-semantics SGD."""
import torch
import torch.nn as nn

from fedtorch_amd.parallel.arena import Arena, ALIGN
from fedtorch_amd.components.optim.sgd import FusedSGD


def small_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Linear(10, 16), nn.BatchNorm1d(16), nn.ReLU(), nn.Linear(16, 4))


def named_model(seed=0):
    """model with 'bn' in some param names like the reference zoos."""
    class M(nn.Module):
        def __init__(self):
            super().__init__()
            torch.manual_seed(seed)
            self.fc1 = nn.Linear(10, 16)
            self.bn1 = nn.BatchNorm1d(16)
            self.fc2 = nn.Linear(16, 4)

        def forward(self, x):
            return self.fc2(torch.relu(self.bn1(self.fc1(x))))
    return M()


def test_arena_views_and_alignment():
    m = named_model()
    arena = Arena(m)
    assert arena.check_views()
    assert arena.numel % ALIGN == 0
    for off in arena.offsets:
        assert off % ALIGN == 0
    # bn params packed at the end, wd prefix excludes them
    bn_numel = sum(p.numel() for n, p in m.named_parameters() if 'bn' in n)
    assert arena.wd_numel < arena.numel
    assert arena.numel - arena.wd_numel >= bn_numel
    # forward/backward accumulate into the grad arena
    x = torch.randn(8, 10)
    m(x).sum().backward()
    assert arena.grad.abs().sum() > 0
    # load/clone round-trip
    snap = arena.clone_flat()
    arena.flat.add_(1.0)
    arena.load_flat(snap)
    for p, v in zip(arena.params, arena.views_of(snap)):
        assert torch.equal(p.data, v)


def reference_sgd_step(params, grads, state, lr=0.1, wd=0.0, in_m=0.0,
                       out_m=0.0, damp=0.0, nesterov=False, apply_lr=True,
                       scale=1.0, apply_in=True, apply_out=False,
                       wd_mask=None):
    """Literal reference semantics (`components/optimizers/sgd.py:81-128`)."""
    for idx, (p, g) in enumerate(zip(params, grads)):
        d_p = g.clone()
        if wd != 0 and apply_lr and (wd_mask is None or wd_mask[idx]):
            d_p.add_(p, alpha=wd)
        if in_m != 0 and apply_in:
            key = ('in', idx)
            if key not in state:
                buf = state[key] = torch.zeros_like(p)
                buf.mul_(in_m).add_(d_p)
            else:
                buf = state[key]
                buf.mul_(in_m).add_(d_p, alpha=1 - damp)
            d_p = d_p.add(buf, alpha=in_m) if nesterov else buf.clone()
        if out_m != 0 and apply_out:
            key = ('out', idx)
            if key not in state:
                buf = state[key] = torch.zeros_like(p)
                buf.mul_(out_m).add_(d_p)
            else:
                buf = state[key]
                buf.mul_(out_m).add_(d_p, alpha=1 - damp)
            d_p = d_p.add(buf, alpha=out_m) if nesterov else buf.clone()
        p.add_(d_p, alpha=-(lr if apply_lr else scale))


def run_both(steps=5, wd=0.01, in_m=0.9, nesterov=False):
    torch.manual_seed(42)
    m1 = named_model(1)
    m2 = named_model(1)
    arena = Arena(m1)
    opt = FusedSGD(arena, lr=0.1, in_momentum=in_m, out_momentum=0.5,
                   weight_decay=wd, nesterov=nesterov)
    ref_params = [p.detach().clone() for _, p in m2.named_parameters()]
    names = [n for n, _ in m2.named_parameters()]
    wd_mask = ['bn' not in n for n in names]
    state = {}
    for s in range(steps):
        torch.manual_seed(100 + s)
        x = torch.randn(8, 10)
        y = torch.randn(8, 4)
        # arena path
        opt.zero_grad()
        ((m1(x) - y) ** 2).mean().backward()
        opt.step(apply_lr=True, apply_in_momentum=in_m != 0,
                 apply_out_momentum=False)
        # reference path: same grads computed on the reference model
        for p_ref, (_, p_live) in zip(ref_params, m2.named_parameters()):
            p_live.data.copy_(p_ref)
        for p in m2.parameters():
            p.grad = None
        ((m2(x) - y) ** 2).mean().backward()
        grads = [p.grad.clone() for p in m2.parameters()]
        reference_sgd_step(ref_params, grads, state, lr=0.1, wd=wd,
                           in_m=in_m, nesterov=nesterov, wd_mask=wd_mask)
    # compare (arena orders decay-first; map by name)
    ref_by_name = dict(zip(names, ref_params))
    for name, p in zip(arena.names, arena.params):
        assert torch.allclose(p.data, ref_by_name[name], atol=1e-6), name


def test_fused_sgd_matches_reference_plain():
    run_both(wd=0.0, in_m=0.0)


def test_fused_sgd_matches_reference_momentum_wd():
    run_both(wd=0.01, in_m=0.9)


def test_fused_sgd_matches_reference_nesterov():
    run_both(wd=0.005, in_m=0.9, nesterov=True)


def test_sync_step_with_out_momentum():
    torch.manual_seed(3)
    m = named_model(2)
    arena = Arena(m)
    opt = FusedSGD(arena, lr=0.1, in_momentum=0.9, out_momentum=0.8)
    agg = torch.randn_like(arena.flat)
    flat0 = arena.clone_flat()
    opt.step(apply_lr=False, scale=0.5, apply_in_momentum=False,
             apply_out_momentum=True, grad=agg)
    # first sync: out buffer = agg, p -= 0.5*agg
    assert torch.allclose(arena.flat, flat0 - 0.5 * agg, atol=1e-6)
    flat1 = arena.clone_flat()
    opt.step(apply_lr=False, scale=0.5, apply_in_momentum=False,
             apply_out_momentum=True, grad=agg)
    # second: buf = 0.8*agg + agg; p -= 0.5*buf
    assert torch.allclose(arena.flat, flat1 - 0.5 * (1.8 * agg), atol=1e-5)


def test_arena_packs_bn_running_stats():
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    m = nn.Sequential(nn.Conv2d(3, 4, 3), nn.BatchNorm2d(4), nn.ReLU())
    arena = Arena(m)
    assert arena.buf_flat is not None
    bn = m[1]
    # views point into the buffer arena
    base = arena.buf_flat.data_ptr()
    end = base + arena.buf_flat.numel() * 4
    assert base <= bn.running_mean.data_ptr() < end
    assert base <= bn.running_var.data_ptr() < end
    # a forward updates stats inside the flat buffer
    m.train()
    m(torch.randn(8, 3, 8, 8))
    assert arena.buf_flat.abs().sum() > 0
    # no-BN model: no buffer arena
    arena2 = Arena(nn.Linear(4, 2))
    assert arena2.buf_flat is None


def test_bnrelu_deepcopy_safe():
    """APFL/PerFedMe deepcopy converted models; the copy's BN must be its
    OWN module (no bound-method aliasing back to the original)."""
    from copy import deepcopy
    from fedtorch_amd.ops.batchnorm import BNReLU, convert_to_fused_bn
    m = nn.Sequential(nn.Conv2d(3, 4, 3), BNReLU(4))
    convert_to_fused_bn(m)
    c = deepcopy(m)
    assert c[1].bn is not m[1].bn
    assert c[1].bn.weight is not m[1].bn.weight
    x = torch.randn(2, 3, 8, 8)
    m.eval()
    c.eval()
    assert torch.allclose(m(x), c(x))
    # relu applied exactly once on the fallback path
    assert (c(x) >= 0).all()


def test_arena_channels_last_packing():
    """4D params of a channels_last module keep channels_last strides
    through the arena, values round-trip, and the model still runs."""
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    torch.manual_seed(3)
    m = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1, bias=False),
                      nn.Conv2d(8, 8, 1, bias=False))
    m = m.to(memory_format=torch.channels_last)
    before = [p.detach().clone() for p in m.parameters()]
    a = Arena(m)
    cl = torch.channels_last
    w0 = m[0].weight
    assert w0.is_contiguous(memory_format=cl) and not w0.is_contiguous()
    for p, b in zip(m.parameters(), before):
        assert torch.equal(p.detach(), b)
    assert a.check_views()
    x = torch.randn(2, 3, 8, 8).contiguous(memory_format=cl)
    y = m(x)
    y.sum().backward()
    # grads accumulated into the arena (attached views)
    assert a.grad.abs().sum() > 0
    # gather fallback path (CPU): detach + eager backward + gather copies
    a.detach_grads()
    a.zero_grad()
    m(x).sum().backward()
    a.gather_grads()
    assert a.grad.abs().sum() > 0


def test_arena_bf16_compute_cpu_roundtrip():
    """enable/disable of the bf16 compute twin re-points params correctly
    (CPU: values only; the fused-kernel path is covered by GPU tests)."""
    import torch.nn as nn
    from fedtorch_amd.parallel.arena import Arena
    torch.manual_seed(4)
    m = nn.Sequential(nn.Linear(8, 8), nn.BatchNorm1d(8))
    a = Arena(m)
    ref = a.flat.clone()
    a.enable_bf16_compute()
    assert m[0].weight.dtype == torch.bfloat16
    assert m[1].weight.dtype == torch.float32  # BN params stay fp32
    assert a.check_views()
    a.flat.mul_(2.0)
    a.sync_half()
    assert torch.allclose(a.half_flat.float(), a.flat, atol=1e-1)
    a.disable_bf16_compute()
    assert m[0].weight.dtype == torch.float32
    assert torch.allclose(a.flat, ref * 2.0)
    assert a.check_views()


def test_stem_conversion_gating():
    """convert_stem swaps only 3x3/s1/p1 Ci=3 Co in {16,32} stems; ImageNet
    7x7 stems and densenet growth stems stay stock Conv2d."""
    from types import SimpleNamespace
    import torch.nn as nn
    from fedtorch_amd.components.model import define_model
    from fedtorch_amd.ops.stemconv import convert_stem, NhwcStemConv

    a = SimpleNamespace(arch='resnet20', data='cifar10', debug=False)
    m = convert_stem(define_model(a))
    assert isinstance(m.conv1, NhwcStemConv)

    a = SimpleNamespace(arch='resnet18', data='imagenet', debug=False)
    m = convert_stem(define_model(a))
    assert type(m.conv1) is nn.Conv2d  # 7x7/s2 stem: not converted

    a = SimpleNamespace(arch='densenet121', data='cifar10', debug=False,
                        densenet_growth_rate=12, densenet_bc_mode=False,
                        densenet_compression=0.5, drop_rate=0.0)
    m = convert_stem(define_model(a))
    assert type(m.conv1) is nn.Conv2d  # Co=24: not a compiled shape
