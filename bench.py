#!/usr/bin/env python3
"""Flagship benchmark: 2x event-stream SR training throughput on MI355X.

Measures the BASELINE.json headline metric — SR event-frames/sec for the
whole job — on the named config: 2x ESR, NFS-syn-shaped synthetic data
(window=2048 events, LR 128x128 -> HR 256x256), seq_len=8 / seqn=3 BPTT,
bf16 autocast, Adam; one full BPTT optimizer step per iteration exactly as
the trainer does (ESR:train_ours_cnt_seq.py:210-235 semantics).

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi-GPU (driver-launched):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist
import torch.nn.functional as F


def synth_sequences(n_seq, seql, seqn, batch, window, lr_res, hr_res, device,
                    seed=0):
    """Pre-generate GPU-resident (inp_scaled_cnt, gt_cnt) window sequences of
    the benchmark shape using the native splat kernels."""
    from esr_amd.data.synthetic import generate_events
    from esr_amd.ops import events_to_channels, normalize_events, \
        scaled_count_encoding
    import numpy as np

    seqs = []
    rng = np.random.default_rng(seed)
    n_windows = seql - seqn + 1
    for s in range(n_seq):
        windows = []
        for w in range(n_windows):
            inp_frames, gt_frames = [], []
            for b in range(batch):
                for f in range(seqn):
                    ev = generate_events(window, lr_res,
                                         seed=int(rng.integers(1 << 30)))
                    ev_t = torch.from_numpy(ev).float()
                    norm = normalize_events(ev_t, lr_res)
                    inp_frames.append(scaled_count_encoding(norm, hr_res, "cnt"))
                    gt_ev = generate_events(window * 4, hr_res,
                                            seed=int(rng.integers(1 << 30)))
                    gt_t = torch.from_numpy(gt_ev).float()
                    gt_frames.append(events_to_channels(gt_t[0], gt_t[1],
                                                        gt_t[3], hr_res))
            inp = torch.stack(inp_frames).view(batch, seqn, 2, *hr_res)
            gt = torch.stack(gt_frames).view(batch, seqn, 2, *hr_res)
            mid = (seqn - 1) // 2
            windows.append((inp.to(device), gt[:, mid].contiguous().to(device)))
        seqs.append(windows)
    return seqs


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=8, help="per-GPU batch")
    p.add_argument("--seql", type=int, default=8)
    p.add_argument("--seqn", type=int, default=3)
    p.add_argument("--window", type=int, default=2048)
    p.add_argument("--lr-size", type=int, default=128)
    p.add_argument("--scale", type=int, default=2)
    p.add_argument("--basech", type=int, default=8)
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32"])
    p.add_argument("--upsampler", type=str, default="pixelshuffle")
    args = p.parse_args()

    assert torch.cuda.is_available(), "bench.py requires an MI355X GPU"

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if world > 1:
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl", init_method="env://")
    device = torch.device(f"cuda:{local_rank}")
    torch.manual_seed(1234 + rank)

    from esr_amd.models import build_model
    from esr_amd.ops.native import require_ext
    require_ext()  # fail loudly if HIP kernels are missing

    model = build_model("ESRNet", inch=2, basech=args.basech,
                        num_frame=args.seqn,
                        upsampler=args.upsampler).to(device)
    if world > 1:
        model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank], bucket_cap_mb=64,
            gradient_as_bucket_view=True)
    inner = model.module if hasattr(model, "module") else model
    optimizer = torch.optim.Adam(model.parameters(), lr=1e-3,
                                 weight_decay=1e-4, amsgrad=True)

    lr_res = (args.lr_size, args.lr_size)
    hr_res = (args.lr_size * args.scale, args.lr_size * args.scale)
    seqs = synth_sequences(2, args.seql, args.seqn, args.batch, args.window,
                           lr_res, hr_res, device, seed=100 + rank)
    n_windows = args.seql - args.seqn + 1
    amp_dtype = torch.bfloat16 if args.dtype == "bf16" else None

    def step(i):
        windows = seqs[i % len(seqs)]
        optimizer.zero_grad(set_to_none=True)
        inner.reset_states()
        loss = 0
        for inp, gt in windows:
            if amp_dtype is not None:
                with torch.autocast("cuda", dtype=amp_dtype):
                    pred = model(inp)
                loss = loss + F.mse_loss(pred.float(), gt)
            else:
                pred = model(inp)
                loss = loss + F.mse_loss(pred, gt)
        loss.backward()
        optimizer.step()
        return loss

    for i in range(args.warmup):
        step(i)

    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    if world > 1:  # max over ranks
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    frames_per_step_per_rank = n_windows * args.batch
    total_frames = frames_per_step_per_rank * args.steps * world
    value = total_frames / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0:
        print(json.dumps({
            "metric": "SR event-frames/sec (whole node), 2x NFS-syn",
            "value": round(value, 2),
            "unit": "frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": f"ESRNet-basech{args.basech}",
                "global_batch": args.batch * world,
                "seq_len": args.seql,
                "seqn": args.seqn,
                "window_events": args.window,
                "input": f"{args.lr_size}->{args.lr_size * args.scale}",
                "scale": args.scale,
                "upsampler": args.upsampler,
                "parallelism": f"dp{world}",
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
