#!/usr/bin/env python3
"""Serving benchmark: StreamingESR per-window latency and throughput.

The serving path is the recurrent pipeline itself (state carried across
windows), hipGraph-replayed per window on GPU.  Reports p50/p95/p99
latency and windows/s for a single stream, and aggregate throughput for
several concurrent streams (one StreamingESR each, shared device).

  python tools/bench_serving.py --windows 200 [--streams 4] [--events]
"""

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def make_window(res, n, seed, device):
    g = torch.Generator().manual_seed(seed)
    H, W = res
    return torch.stack([
        torch.randint(0, W, (n,), generator=g).float(),
        torch.randint(0, H, (n,), generator=g).float(),
        torch.sort(torch.rand(n, generator=g)).values,
        torch.randint(0, 2, (n,), generator=g).float() * 2 - 1]).to(device)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--windows", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=20)
    ap.add_argument("--streams", type=int, default=1)
    ap.add_argument("--lr-size", type=int, default=128)
    ap.add_argument("--scale", type=int, default=2)
    ap.add_argument("--window-events", type=int, default=2048)
    ap.add_argument("--basech", type=int, default=8)
    ap.add_argument("--events", action="store_true",
                    help="emit HR event streams (push_events) instead of "
                         "count maps")
    ap.add_argument("--device", default="cuda:0"
                    if torch.cuda.is_available() else "cpu")
    args = ap.parse_args()

    from esr_amd.engine.streaming import StreamingESR
    from esr_amd.models import build_model

    device = torch.device(args.device)
    torch.manual_seed(0)
    res = (args.lr_size, args.lr_size)
    import copy
    model = build_model("ESRNet", inch=2, basech=args.basech, num_frame=3,
                        upsampler="pixelshuffle")
    # one model copy per stream: the recurrent state lives in the model,
    # so concurrent streams must not share it
    servers = [StreamingESR(copy.deepcopy(model), res, scale=args.scale,
                            seqn=3, device=device,
                            use_graphs=device.type == "cuda")
               for _ in range(args.streams)]
    windows = [make_window(res, args.window_events, i, device)
               for i in range(16)]
    cap = args.window_events * args.scale ** 2

    def push(srv, w):
        if args.events:
            return srv.push_events(w, capacity=cap)
        return srv.push(w)

    for i in range(args.warmup):
        for srv in servers:
            push(srv, windows[i % len(windows)])
    if device.type == "cuda":
        torch.cuda.synchronize()

    lat = []
    t0 = time.perf_counter()
    for i in range(args.windows):
        tw = time.perf_counter()
        for srv in servers:
            out = push(srv, windows[i % len(windows)])
        if device.type == "cuda":
            torch.cuda.synchronize()
        lat.append(time.perf_counter() - tw)
    total = time.perf_counter() - t0

    lat_ms = sorted(v * 1e3 for v in lat)

    def pct(p):
        return lat_ms[min(len(lat_ms) - 1, int(p / 100 * len(lat_ms)))]

    import json
    print(json.dumps({
        "metric": "serving windows/s",
        "value": round(args.windows * args.streams / total, 2),
        "streams": args.streams,
        "latency_ms": {"p50": round(pct(50), 3), "p95": round(pct(95), 3),
                       "p99": round(pct(99), 3)},
        "emit": "hr_event_stream" if args.events else "hr_count_map",
        "config": {"lr": args.lr_size, "scale": args.scale,
                   "window_events": args.window_events,
                   "graphs": servers[0].use_graphs},
    }))


if __name__ == "__main__":
    main()
