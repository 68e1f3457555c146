#!/usr/bin/env python3
"""Bisect which op breaks hipGraph capture (GPU box debugging tool)."""

import sys
import traceback
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from esr_amd.models import build_model
from esr_amd.ops.native import require_ext

require_ext()
dev = torch.device("cuda:0")


def try_capture(name, warmup_fn, capture_fn):
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                warmup_fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            capture_fn()
        torch.cuda.synchronize()
        g.replay()
        torch.cuda.synchronize()
        print(f"[OK]   {name}")
        return True
    except Exception as e:
        print(f"[FAIL] {name}: {type(e).__name__}: {e}")
        traceback.print_exc()
        return False


# stage 1: plain conv fwd
m1 = torch.nn.Conv2d(2, 8, 3, padding=1).to(dev)
x1 = torch.randn(8, 2, 64, 64, device=dev)
try_capture("conv fwd", lambda: m1(x1), lambda: m1(x1))

# stage 2: ESRNet fwd fp32
m = build_model("ESRNet", inch=2, basech=8, num_frame=3,
                upsampler="pixelshuffle").to(dev)
x = torch.randn(2, 3, 2, 128, 128, device=dev)


def fwd():
    m.reset_states()
    return m(x)


try_capture("ESRNet fwd fp32", fwd, fwd)

# stage 2b: submodules isolated
head = m.head
fe = m.feat_extract
tp = m.time_propagate
sf = m.spacetime_fuse
xf = torch.randn(6, 2, 128, 128, device=dev)
try_capture("head+encoder", lambda: fe(head(xf)), lambda: fe(head(xf)))
deep = torch.randn(2, 3, 64, 16, 16, device=dev)


def run_tp():
    tp.reset_states()
    return tp(deep)


try_capture("time_propagate", run_tp, run_tp)
feats_list = [torch.randn(6, 64, 16, 16, device=dev),
              torch.randn(6, 32, 32, 32, device=dev),
              torch.randn(6, 16, 64, 64, device=dev)]
try_capture("stfusion", lambda: sf(deep, feats_list),
            lambda: sf(deep, feats_list))

# stage 3: fwd+bwd fp32
params = [p for p in m.parameters() if p.requires_grad]
flat = torch.zeros(sum(p.numel() for p in params), device=dev)
off = 0
for p in params:
    p.grad = flat[off:off + p.numel()].view_as(p)
    off += p.numel()


def fwd_bwd():
    flat.zero_()
    m.reset_states()
    loss = (m(x) ** 2).mean()
    loss.backward()


try_capture("ESRNet fwd+bwd fp32", fwd_bwd, fwd_bwd)

# stage 4: + capturable Adam
opt = torch.optim.Adam(params, lr=1e-3, weight_decay=1e-4, amsgrad=True,
                       foreach=True, capturable=True)


def full():
    fwd_bwd()
    opt.step()


try_capture("fwd+bwd+Adam", full, full)


# stage 5: bf16 autocast fwd+bwd
def fwd_bwd_amp():
    flat.zero_()
    m.reset_states()
    with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
        y = m(x)
    (y.float() ** 2).mean().backward()


try_capture("fwd+bwd bf16 autocast", fwd_bwd_amp, fwd_bwd_amp)
print("done")
