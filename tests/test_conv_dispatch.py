"""CPU-side tests for the native-conv dispatch layer (ops/conv.py):
weight packing layout and the shape-dispatch predicate — the halves of
the GPU path that are pure host logic."""

import torch
import torch.nn as nn

from esr_amd.ops.conv import _pack, conv2d_act, shape_supported


def test_pack_layout_matches_fragment_addressing():
    """_pack must produce [tap][Cout_p][Cin_p] with zero padding — the
    exact layout conv2d_fwd_mfma_kernel's A-fragment addressing assumes
    (row = cout, contiguous 8-channel groups along ci)."""
    g = torch.Generator().manual_seed(0)
    O, I, k = 33, 40, 3
    w = torch.randn(O, I, k, k, generator=g).to(torch.bfloat16)
    p = _pack(w)
    assert p.shape == (9, 48, 64)            # ceil16(33), ceil32(40)
    assert p.is_contiguous()
    for tap in [0, 4, 8]:
        ky, kx = tap // 3, tap % 3
        assert torch.equal(p[tap, :O, :I], w[:, :, ky, kx])
    assert not p[:, O:, :].any() and not p[:, :, I:].any()


def test_pack_no_padding_needed():
    w = torch.randn(64, 64, 1, 1).to(torch.bfloat16)
    p = _pack(w)
    assert p.shape == (1, 64, 64)
    assert torch.equal(p[0], w[:, :, 0, 0])


def test_shape_dispatch_table():
    """The measured per-shape routing rules (profiles/README.md)."""
    def conv(cin, cout, k, s, p=None, groups=1, dilation=1):
        return nn.Conv2d(cin, cout, k, s, k // 2 if p is None else p,
                         dilation=dilation, groups=groups)

    # supported: every conv the flagship model runs
    assert shape_supported(conv(2, 8, 3, 1))        # head (valu2)
    assert shape_supported(conv(8, 16, 3, 2)) is False  # enc1 s2 -> MIOpen
    assert shape_supported(conv(16, 32, 3, 2))      # enc2 (MFMA)
    assert shape_supported(conv(192, 192, 3, 1))    # resblock
    assert shape_supported(conv(128, 64, 1, 1))     # 1x1 fusion
    assert shape_supported(conv(64, 1, 3, 1))       # attention map (valu)
    assert shape_supported(conv(8, 2, 3, 1))        # tail (valu2)
    # unsupported configurations fall back to torch
    assert not shape_supported(conv(64, 64, 5, 1))          # 5x5
    assert not shape_supported(conv(64, 64, 3, 1, p=0))     # pad != k//2
    assert not shape_supported(conv(64, 64, 3, 1, groups=2))
    assert not shape_supported(conv(64, 64, 3, 1, dilation=2))
    assert not shape_supported(nn.Conv2d(64, 64, (3, 1), 1, (1, 0)))
    assert not shape_supported(conv(1, 1, 1, 2, p=0))       # 1x1 s2


def test_conv2d_act_cpu_returns_none():
    """On CPU the dispatcher must decline so callers use torch — the CPU
    suite never silently depends on the HIP extension."""
    layer = nn.Conv2d(64, 64, 3, 1, 1)
    x = torch.randn(1, 64, 16, 16)
    assert conv2d_act(x, layer, "relu") is None


def test_conv2d_act_rejects_unknown_activation():
    layer = nn.Conv2d(64, 64, 3, 1, 1)
    x = torch.randn(1, 64, 16, 16)
    assert conv2d_act(x, layer, "gelu") is None
