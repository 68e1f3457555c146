#!/usr/bin/env python3
"""Microbenchmark: native gfx950 conv kernels vs torch/MIOpen per shape.

Runs the conv shapes ESRNet's flagship config actually executes (batch
sizes as in the batch-64 bench step) and prints native vs torch (bf16,
NCHW) forward and fwd+bwd times.

Usage (GPU box): python tools/bench_conv.py [--iters 50]
"""

import argparse
import sys
import time
from pathlib import Path

import torch
import torch.nn.functional as F

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

# (label, frames, Cin, Cout, H, W, ks, stride, act) — flagship step shapes
SHAPES = [
    ("head 2->8 @256", 512, 2, 8, 256, 256, 3, 1, "relu"),
    ("enc1 8->16 s2 @256", 512, 8, 16, 256, 256, 3, 2, "relu"),
    ("enc2 16->32 s2 @128", 512, 16, 32, 128, 128, 3, 2, "relu"),
    ("enc3 32->64 s2 @64", 512, 32, 64, 64, 64, 3, 2, "relu"),
    ("ltc predmap 128->64 @32", 768, 128, 64, 32, 32, 3, 1, "relu"),
    ("ltc predmap 64->1 @32", 768, 64, 1, 32, 32, 3, 1, "sigmoid"),
    ("ltc resblock 192->192 @32", 384, 192, 192, 32, 32, 3, 1, "relu"),
    ("ltc fuse 192->64 @32", 384, 192, 64, 32, 32, 3, 1, None),
    ("gru ur 128->128 @32", 768, 128, 128, 32, 32, 3, 1, None),
    ("gru out 128->64 @32", 768, 128, 64, 32, 32, 3, 1, None),
    ("gfuse 128->64 1x1 @32", 384, 128, 64, 32, 32, 1, 1, "relu"),
    ("fusion 128->64 @32", 768, 128, 64, 32, 32, 3, 1, "relu"),
    ("dense 192->64 @32", 384, 192, 64, 32, 32, 3, 1, "relu"),
    ("recon1 64->128 @32", 384, 64, 128, 32, 32, 3, 1, None),
    ("recon2 32->64 @64", 384, 32, 64, 64, 64, 3, 1, None),
    ("recon3 16->32 @128", 384, 16, 32, 128, 128, 3, 1, None),
    ("tail 8->2 @256", 384, 8, 2, 256, 256, 3, 1, "relu"),
]


def bench(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def valu2_component(args):
    """Tiny-channel shapes: v2 strip kernel vs v1 VALU vs torch/MIOpen."""
    from esr_amd.ops.conv import ACT_IDS
    from esr_amd.ops.native import require_ext
    ext = require_ext()
    dev = "cuda:0"
    torch.manual_seed(0)
    tiny = [sh for sh in SHAPES if sh[2] < 32 and sh[3] < 32]
    print(f"{'shape':34s} {'v2 ms':>9s} {'v1 ms':>9s} {'torch':>9s} {'t/v2':>6s}")
    for label, B, cin, cout, h, w, ks, stride, act in tiny:
        x = torch.randn(B, cin, h, w, device=dev).to(torch.bfloat16)
        wt = (torch.randn(cout, cin, ks, ks, device=dev) * 0.2) \
            .to(torch.bfloat16)
        b = torch.randn(cout, device=dev).float()
        aid = ACT_IDS[act]
        act_fn = {None: lambda t: t, "relu": F.relu,
                  "sigmoid": torch.sigmoid, "tanh": torch.tanh}[act]

        def v2():
            return ext.conv2d_fwd_valu2(x, wt, b, stride, aid)

        def v1():
            return ext.conv2d_fwd_valu(x, wt, b, stride, aid)

        def ref():
            return act_fn(F.conv2d(x, wt, b.to(torch.bfloat16),
                                   stride=stride, padding=ks // 2))
        # correctness spot check before timing
        got = v2().float()
        want = act_fn(F.conv2d(x.float(), wt.float(), b,
                               stride=stride, padding=ks // 2))
        err = (got - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 2e-2, f"{label}: v2 wrong, rel {err / scale}"
        t2 = bench(v2, args.iters)
        t1 = bench(v1, args.iters)
        tt = bench(ref, args.iters)
        print(f"{label:34s} {t2:9.3f} {t1:9.3f} {tt:9.3f} {tt / t2:6.2f}")


def wgrad_component(args):
    """Isolated weight-grad: native conv2d_wgrad_mfma vs aten wgrad-only."""
    from esr_amd.ops.native import require_ext
    ext = require_ext()
    dev = "cuda:0"
    torch.manual_seed(0)

    def ceil(v, m):
        return (v + m - 1) // m * m

    print(f"{'shape':34s} {'native ms':>10s} {'aten ms':>10s} {'x':>6s}")
    for label, B, cin, cout, h, w, ks, stride, _ in SHAPES:
        x = torch.randn(B, cin, h, w, device=dev).to(torch.bfloat16)
        dy_h = (h + 2 * (ks // 2) - ks) // stride + 1
        dy_w = (w + 2 * (ks // 2) - ks) // stride + 1
        dy = torch.randn(B, cout, dy_h, dy_w, device=dev).to(torch.bfloat16)
        wt = torch.randn(cout, cin, ks, ks, device=dev).to(torch.bfloat16)

        def native():
            return ext.conv2d_wgrad_mfma(x, dy, ks, stride,
                                         ceil(cin, 16), ceil(cout, 16))

        def aten():
            return torch.ops.aten.convolution_backward(
                dy, x, wt, None, [stride, stride], [ks // 2, ks // 2],
                [1, 1], False, [0, 0], 1, [False, True, False])[1]

        tn = bench(native, args.iters)
        tt = bench(aten, args.iters)
        print(f"{label:34s} {tn:10.3f} {tt:10.3f} {tt / tn:6.2f}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--bwd", action="store_true", help="also time fwd+bwd")
    ap.add_argument("--wgrad", action="store_true",
                    help="time the wgrad kernel alone vs aten wgrad-only")
    ap.add_argument("--valu2", action="store_true",
                    help="A/B the tiny-channel strip kernel vs v1 and torch")
    args = ap.parse_args()

    if args.wgrad:
        return wgrad_component(args)
    if args.valu2:
        return valu2_component(args)

    from esr_amd.ops.conv import ACT_IDS, _NativeConv2dFn
    dev = "cuda:0"
    torch.manual_seed(0)

    total_n, total_t = 0.0, 0.0
    print(f"{'shape':34s} {'native ms':>10s} {'torch ms':>10s} {'x':>6s}")
    for label, B, cin, cout, h, w, ks, stride, act in SHAPES:
        x = torch.randn(B, cin, h, w, device=dev).to(torch.bfloat16)
        wt = (torch.randn(cout, cin, ks, ks, device=dev) * 0.2) \
            .to(torch.bfloat16)
        b = torch.randn(cout, device=dev).to(torch.bfloat16)
        act_fn = {None: lambda t: t, "relu": F.relu,
                  "sigmoid": torch.sigmoid, "tanh": torch.tanh}[act]

        def native():
            return _NativeConv2dFn.apply(x, wt, b, stride, ACT_IDS[act])

        def ref():
            return act_fn(F.conv2d(x, wt, b, stride=stride, padding=ks // 2))

        if args.bwd:
            xg = x.clone().requires_grad_(True)
            wg = wt.clone().requires_grad_(True)
            bgr = b.clone().requires_grad_(True)
            y0 = native()
            gy = torch.randn_like(y0)

            def native_b():
                y = _NativeConv2dFn.apply(xg, wg, bgr, stride, ACT_IDS[act])
                y.backward(gy)

            xr = x.clone().requires_grad_(True)
            wr = wt.clone().requires_grad_(True)
            br = b.clone().requires_grad_(True)

            def ref_b():
                y = act_fn(F.conv2d(xr, wr, br, stride=stride,
                                    padding=ks // 2))
                y.backward(gy)
            tn = bench(native_b, args.iters)
            tt = bench(ref_b, args.iters)
        else:
            tn = bench(native, args.iters)
            tt = bench(ref, args.iters)
        total_n += tn
        total_t += tt
        print(f"{label:34s} {tn:10.3f} {tt:10.3f} {tt / tn:6.2f}")
    print(f"{'TOTAL':34s} {total_n:10.3f} {total_t:10.3f} "
          f"{total_t / total_n:6.2f}")


if __name__ == "__main__":
    main()
