"""Native bf16 NCHW convolution dispatch (hand-written gfx950 kernels).

Replaces MIOpen + NCHW<->NHWC transposes + autocast casts on the conv
shapes ESRNet runs (reference ConvLayer ESR:models/submodules.py:159-200,
ConvGRU convs :474-514): 3x3 stride-1/2 pad-1 and 1x1 convs, bias and
ReLU/sigmoid/tanh fused into the conv epilogue.

Dispatch (see conv2d.hip; thresholds from tools/bench_conv.py measured on
MI355X, profiles/README.md):
  * MFMA implicit-GEMM kernel for Cout >= 32 — measured 1.0-2.0x MIOpen
    bf16 on the deep/mid shapes;
  * direct VALU kernel for Cin >= 32 with tiny Cout (attention/kernel
    convs, 1.2-1.3x);
  * register-strip VALU v2 for the tiny-channel class (Cin and Cout both
    < 32): 8-wide output strips x all cout per thread, weights broadcast
    from LDS — head 2.7x, tail 1.3x vs MIOpen; the stride-2 enc1 shape
    (0.67x) stays on MIOpen.
Backward defaults to torch's aten.convolution_backward (MIOpen bf16).
The native backward kernels are implemented, oracle-tested and measured
(tools/bench_conv.py --wgrad / --bwd, MI355X):
  * weight-grad (conv2d_wgrad_s1: 4-row MFMA chunks, rolling X window,
    DETERMINISTIC two-stage reduction — no fp32 atomics) reaches
    0.9-1.1x MIOpen wrw on the wide 32x32 shapes (resblock 1.07x,
    192->64 1.05x, dense 1.03x) after a 7x optimization ladder driven by
    ablation (the original per-address atomic flush was 70-87% of the
    kernel; ESR_WGRAD_ABL still exposes the ablation arms);
  * stride-1 input-grad reuses the forward kernels with packed
    flipped/transposed weights; stride-2 dgrad is a VALU gather kernel.
End-to-end the native backward composition still measures 0.8x of
MIOpen's fused bwd on the whole shape set, so aten stays the default;
ESR_CONV_BWD=native forces all-native, =auto uses native for deep
stride-1 shapes.  fp32 falls back to torch everywhere (the CPU oracle of
the parity tests).

Env: ESR_NATIVE_CONV=0 disables the native path (A/B benching);
     ESR_CONV_BWD=native routes backward to the native kernels.
"""

from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from .native import get_ext

__all__ = ["conv2d_act", "native_conv_supported", "ACT_IDS"]

ACT_IDS = {None: 0, "none": 0, "relu": 1, "sigmoid": 2, "tanh": 3}

# dispatch telemetry: tests / profiling assert the native path really runs
stats = {"native_calls": 0}

_MFMA_MIN_COUT = 32   # below this the MFMA M-tile would idle; use VALU
_CIK = 32             # K-chunk of the MFMA kernel (pad Cin up to this)


def _enabled() -> bool:
    return os.environ.get("ESR_NATIVE_CONV", "1") != "0"


def _ceil(v: int, m: int) -> int:
    return (v + m - 1) // m * m


def _pack(w: torch.Tensor) -> torch.Tensor:
    """[O, I, k, k] -> [k*k, O_p, I_p] bf16, O_p = ceil16(O), I_p = ceil32(I).

    Matches the A-fragment addressing of conv2d_fwd_mfma_kernel: row = cout,
    contiguous 8-channel groups along ci.
    """
    O, I, k, _ = w.shape
    t = w.permute(2, 3, 0, 1).reshape(k * k, O, I)
    op, ip = _ceil(O, 16), _ceil(I, _CIK)
    if op != O or ip != I:
        t = F.pad(t, (0, ip - I, 0, op - O))
    return t.contiguous()


class _NativeConv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, act_id):
        ext = get_ext()
        ks = w.shape[2]
        cout, cin = w.shape[0], w.shape[1]
        bias_f = bias.float() if bias is not None else None
        if cout >= _MFMA_MIN_COUT:
            y = ext.conv2d_fwd_mfma(x, _pack(w), bias_f, cout, ks, stride,
                                    act_id)
        elif cin < _MFMA_MIN_COUT and x.shape[-1] % 8 == 0:
            # tiny-channel head/tail class: register-strip kernel
            # (2.7x / 1.3x MIOpen on head/tail, bench_conv --valu2).
            # Widths not divisible by 8 conservatively take the v1 kernel
            # (their strip stores are row-misaligned; the suspected odd-W
            # failure later turned out to be a test-harness fp32-mean
            # artifact, but v2 at odd W is GPU-unverified) — every real
            # model shape is /8-padded anyway (ESRNet.DOWN_SCALE).
            y = ext.conv2d_fwd_valu2(x, w.contiguous(), bias_f, stride,
                                     act_id)
        else:
            y = ext.conv2d_fwd_valu(x, w.contiguous(), bias_f, stride, act_id)
        ctx.save_for_backward(x, w, y)
        ctx.stride_ = stride
        ctx.act_id = act_id
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext()
        x, w, y = ctx.saved_tensors
        stride, act_id = ctx.stride_, ctx.act_id
        ks = w.shape[2]
        cout, cin = w.shape[0], w.shape[1]
        dy = dy.contiguous()
        dpre = ext.act_grad(dy, y, act_id) if act_id else dy

        # default aten: the native backward kernels (v2) are oracle-correct
        # but still measured behind MIOpen's wrw/bwd igemm on the deep
        # shapes (tools/bench_conv.py --bwd); "auto" switches stride-1
        # deep shapes native once they win
        mode = os.environ.get("ESR_CONV_BWD", "aten")
        native_bwd = mode == "native" or (
            mode == "auto" and stride == 1 and cin >= _MFMA_MIN_COUT)
        if not native_bwd:
            dx, dw, db = torch.ops.aten.convolution_backward(
                dpre, x, w, [cout] if ctx.has_bias else None,
                [stride, stride], [ks // 2, ks // 2], [1, 1], False, [0, 0],
                1, [True, True, ctx.has_bias])
            return dx, dw, db if ctx.has_bias else None, None, None

        # --- native backward path (ESR_CONV_BWD=native) ---
        # input grad: stride-1 is this conv with flipped/transposed weights
        if stride == 1:
            wt = w.transpose(0, 1)
            if ks == 3:
                wt = wt.flip((2, 3))
            if cin >= _MFMA_MIN_COUT:
                dx = ext.conv2d_fwd_mfma(dpre, _pack(wt), None, cin, ks, 1, 0)
            else:
                dx = ext.conv2d_fwd_valu(dpre, wt.contiguous(), None, 1, 0)
        else:
            dx = ext.conv2d_dgrad_s2(dpre, w.contiguous(),
                                     x.shape[2], x.shape[3])

        # weight grad: MFMA split-K into padded fp32 (transposed
        # [Cin_p, Cout_p] scratch for line-parallel atomics), then slice
        dwp = ext.conv2d_wgrad_mfma(x, dpre, ks, stride,
                                    _ceil(cin, 16), _ceil(cout, 16))
        dw = dwp.permute(1, 0, 2, 3)[:cout, :cin].to(w.dtype)
        db = dpre.sum(dim=(0, 2, 3), dtype=torch.float32).to(w.dtype) \
            if ctx.has_bias else None
        return dx, dw, db, None, None


def shape_supported(conv: torch.nn.Conv2d) -> bool:
    """Shape half of the dispatch predicate (CPU-testable)."""
    if conv.out_channels < _MFMA_MIN_COUT and conv.in_channels < _MFMA_MIN_COUT:
        # tiny-channel class: the v2 register-strip kernel beats MIOpen at
        # stride 1 (head 2.7x, tail 1.3x); the stride-2 enc1 shape still
        # loses (0.67x) and stays on MIOpen (tools/bench_conv.py --valu2)
        if conv.stride[0] != 1:
            return False
    if conv.groups != 1 or conv.dilation != (1, 1):
        return False
    kh, kw = conv.kernel_size
    sh, sw = conv.stride
    ph, pw = conv.padding if isinstance(conv.padding, tuple) else \
        (conv.padding, conv.padding)
    if kh != kw or sh != sw or ph != pw:
        return False
    if (kh, sh) not in ((3, 1), (3, 2), (1, 1)):
        return False
    if ph != kh // 2:
        return False
    return True


def native_conv_supported(x: torch.Tensor, conv: torch.nn.Conv2d) -> bool:
    if not (_enabled() and x.is_cuda and get_ext() is not None):
        return False
    return shape_supported(conv)


def conv2d_act(x: torch.Tensor, conv: torch.nn.Conv2d, act: str | None):
    """Fused conv+bias+act.  Returns the native-kernel result when the shape
    and dtype qualify, else None (caller falls back to torch)."""
    if act not in ACT_IDS or not native_conv_supported(x, conv):
        return None
    if torch.is_autocast_enabled():
        if torch.get_autocast_dtype("cuda") != torch.bfloat16:
            return None   # fp16 autocast: kernels are bf16-native; fall back
        x = x.to(torch.bfloat16)
    if x.dtype != torch.bfloat16:
        return None
    w, b = conv.weight, conv.bias
    if w.dtype != torch.bfloat16:
        w = w.to(torch.bfloat16)          # autograd records the cast:
        if b is not None:                  # grads flow back to fp32 masters
            b = b.to(torch.bfloat16)
    stats["native_calls"] += 1
    return _NativeConv2dFn.apply(x.contiguous(), w, b, conv.stride[0],
                                 ACT_IDS[act])
