"""Deformable-conv tests (methodology parity: ESR:models/DCNv2/testcpu.py —
zero-offset equivalence with standard conv, and gradcheck in fp64)."""

import torch
import torch.nn.functional as F

from esr_amd.ops.dcn import (DeformAlign2d, _deform_conv2d_torch,
                             modulated_deform_conv2d)


def _rand_problem(B=2, C=4, H=7, W=9, Cout=6, dg=2, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    kw = dict(dtype=dtype)
    input = torch.randn(B, C, H, W, generator=g, **kw)
    offset = torch.randn(B, dg * 2 * 9, H, W, generator=g, **kw) * 2
    mask = torch.rand(B, dg * 9, H, W, generator=g, **kw)
    weight = torch.randn(Cout, C, 3, 3, generator=g, **kw) * 0.2
    bias = torch.randn(Cout, generator=g, **kw)
    return input, offset, mask, weight, bias


def test_zero_offset_equals_conv():
    """offset=0, mask=1 => plain 3x3 conv (ESR:models/DCNv2/testcpu.py:32-67)."""
    input, _, _, weight, bias = _rand_problem()
    offset = torch.zeros(2, 2 * 2 * 9, 7, 9)
    mask = torch.ones(2, 2 * 9, 7, 9)
    out = _deform_conv2d_torch(input, offset, mask, weight, bias,
                               (1, 1), (1, 1), (1, 1), 2)
    ref = F.conv2d(input, weight, bias, 1, 1)
    assert torch.allclose(out, ref, atol=1e-5), \
        (out - ref).abs().max().item()


def test_integer_offset_shifts():
    """An integer offset samples the shifted pixel exactly."""
    input = torch.arange(25.0).view(1, 1, 5, 5)
    weight = torch.zeros(1, 1, 3, 3)
    weight[0, 0, 1, 1] = 1.0          # identity kernel (centre tap only)
    offset = torch.zeros(1, 18, 5, 5)
    offset[:, 2 * 4 + 0] = 1.0        # centre tap (k=4): +1 in H
    mask = torch.ones(1, 9, 5, 5)
    out = _deform_conv2d_torch(input, offset, mask, weight, None,
                               (1, 1), (1, 1), (1, 1), 1)
    # out(y,x) = input(y+1, x) inside; bottom row samples out of range -> 0
    assert torch.allclose(out[0, 0, :4], input[0, 0, 1:])
    assert (out[0, 0, 4] == 0).all()


def test_gradcheck_torch_ref():
    torch.manual_seed(1)
    input = torch.randn(1, 2, 5, 5, dtype=torch.float64, requires_grad=True)
    offset = torch.randn(1, 2 * 1 * 9, 5, 5, dtype=torch.float64,
                         requires_grad=True) * 0.7
    offset.retain_grad()
    mask = torch.rand(1, 9, 5, 5, dtype=torch.float64, requires_grad=True)
    weight = torch.randn(3, 2, 3, 3, dtype=torch.float64, requires_grad=True)
    bias = torch.randn(3, dtype=torch.float64, requires_grad=True)

    def fn(i, o, m, w, b):
        return _deform_conv2d_torch(i, o, m, w, b, (1, 1), (1, 1), (1, 1), 1)

    assert torch.autograd.gradcheck(fn, (input, offset, mask, weight, bias),
                                    eps=1e-6, atol=1e-4)


def test_deform_align_module_init_behavior():
    """Zero-init offset conv => DCN == 0.5 * standard conv at init
    (sigmoid(0) mask), matching DCN_sep (ESR:models/DCNv2/dcn_v2.py:210-227)."""
    torch.manual_seed(2)
    m = DeformAlign2d(4, 4, 3, stride=1, padding=1, deformable_groups=2)
    x = torch.randn(1, 4, 8, 8)
    feat = torch.randn(1, 4, 8, 8)
    out = m(x, feat)
    ref = 0.5 * F.conv2d(x, m.weight, m.bias * 0, 1, 1) + m.bias.view(1, -1, 1, 1)
    assert torch.allclose(out, ref, atol=1e-5)


def test_functional_dispatch_cpu():
    input, offset, mask, weight, bias = _rand_problem()
    out = modulated_deform_conv2d(input, offset, mask, weight, bias,
                                  stride=1, padding=1, dilation=1,
                                  deformable_groups=2)
    assert out.shape == (2, 6, 7, 9)
