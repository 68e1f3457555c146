#!/usr/bin/env python3
"""Microbenchmark the DCN kernels at the flagship shape (GPU box).

  ESR_DCN_TILED=0 python tools/bench_dcn.py   # per-contribution col2im
  python tools/bench_dcn.py                   # LDS-tiled col2im
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from esr_amd.ops.native import require_ext  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    e1.synchronize()
    return e0.elapsed_time(e1) / iters


def main():
    ext = require_ext()
    g = torch.Generator().manual_seed(0)
    B, C, H, W, Cout, dg = 128, 64, 32, 32, 64, 8   # flagship dense_fuse shape
    input = torch.randn(B, C, H, W, generator=g).cuda()
    offset = (torch.randn(B, dg * 18, H, W, generator=g) * 1.5).cuda()
    mask = torch.rand(B, dg * 9, H, W, generator=g).cuda()
    weight = (torch.randn(Cout, C, 3, 3, generator=g) * 0.2).cuda()
    bias = torch.randn(Cout, generator=g).cuda()
    gout = torch.randn(B, Cout, H, W, generator=g).cuda()

    t_fwd_fused = timeit(lambda: ext.deform_conv2d_forward_fused(
        input, offset, mask, weight, bias, dg))
    cols_fn = lambda: ext.deform_im2col(input, offset, mask, 3, 3, 1, 1,  # noqa
                                        1, 1, 1, 1, dg)
    t_im2col = timeit(cols_fn)
    cols = cols_fn()
    t_gemm = timeit(lambda: torch.matmul(weight.reshape(Cout, -1), cols))
    t_bwd = timeit(lambda: ext.deform_conv2d_backward(
        input, offset, mask, weight, gout, 1, 1, 1, 1, 1, 1, dg), iters=20)

    import os
    print(f"ESR_DCN_TILED={os.environ.get('ESR_DCN_TILED', '1')}")
    print(f"fwd fused      : {t_fwd_fused:.3f} ms")
    print(f"fwd im2col     : {t_im2col:.3f} ms (+GEMM {t_gemm:.3f} ms)")
    print(f"bwd (full)     : {t_bwd:.3f} ms")


if __name__ == "__main__":
    main()
