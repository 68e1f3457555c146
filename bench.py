#!/usr/bin/env python3
"""Flagship benchmark: 2x event-stream SR training throughput on MI355X.

Measures the BASELINE.json headline metric — SR event-frames/sec for the
whole job — on the named config: 2x ESR, NFS-syn-shaped synthetic data
(window=2048 events per LR frame, LR 128x128 -> HR 256x256), seq_len=8 /
seqn=3 BPTT, Adam; one full BPTT optimizer step per iteration exactly as
the trainer does (ESR:train_ours_cnt_seq.py:210-235 semantics).

MI355X design: the model is tiny (1.8 M params, ~130 kernels per window),
so a step is launch-latency-bound — the whole BPTT step (6 windows of
forward + one backward + Adam) is captured in a hipGraph and replayed;
fresh synthetic data is copied into the graph's static input buffers every
step.  Synthetic events are generated and splatted ON DEVICE with the
native splat kernels.  --no-graphs falls back to eager.

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


def make_sequence_pool(n_sets, seql, n_windows, batch, seqn, window, lr_res,
                       hr_res, device, ext, seed):
    """Generate pools of (frames [B,seql,2,kH,kW], gts: n_windows x
    [B,2,kH,kW]) per step, fully on device via the native splat kernels.
    A BPTT step has seql UNIQUE frames; the seqn-windows are views."""
    g = torch.Generator(device=device).manual_seed(seed)
    H, W = lr_res
    kH, kW = hr_res
    scale = kH // H
    pools = []
    for _ in range(n_sets):
        BF = batch * seql
        ev = torch.empty(BF, window, 4, device=device)
        ev[..., 0] = (torch.rand(BF, window, device=device, generator=g)
                      * W).floor() * scale
        ev[..., 1] = (torch.rand(BF, window, device=device, generator=g)
                      * H).floor() * scale
        ev[..., 2] = torch.rand(BF, window, device=device, generator=g)
        ev[..., 3] = torch.randint(0, 2, (BF, window), device=device,
                                   generator=g).float() * 2 - 1
        frames = ext.splat_count(ev.contiguous(), kH, kW) \
            .view(batch, seql, 2, kH, kW)
        gts = []
        ng = window * scale * scale
        for _ in range(n_windows):
            gt_ev = torch.empty(batch, ng, 4, device=device)
            gt_ev[..., 0] = (torch.rand(batch, ng, device=device,
                                        generator=g) * kW).floor()
            gt_ev[..., 1] = (torch.rand(batch, ng, device=device,
                                        generator=g) * kH).floor()
            gt_ev[..., 2] = torch.rand(batch, ng, device=device, generator=g)
            gt_ev[..., 3] = torch.randint(0, 2, (batch, ng), device=device,
                                          generator=g).float() * 2 - 1
            gts.append(ext.splat_count(gt_ev.contiguous(), kH, kW))
        pools.append((frames, gts))
    return pools


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=64, help="per-GPU batch")
    p.add_argument("--seql", type=int, default=8)
    p.add_argument("--seqn", type=int, default=3)
    p.add_argument("--window", type=int, default=2048)
    p.add_argument("--lr-size", type=int, default=128)
    p.add_argument("--lr-h", type=int, default=None,
                   help="LR height (defaults to --lr-size; for DVS-native "
                        "resolutions like 180x240)")
    p.add_argument("--lr-w", type=int, default=None)
    p.add_argument("--scale", type=int, default=2)
    p.add_argument("--basech", type=int, default=8)
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp16", "fp32"])
    p.add_argument("--metric-suffix", type=str, default=None,
                   help="override the config label in the JSON line "
                        "(BASELINE config arms)")
    p.add_argument("--upsampler", type=str, default="pixelshuffle")
    p.add_argument("--no-graphs", action="store_true",
                   help="disable hipGraph capture (eager mode)")
    p.add_argument("--channels-last", action="store_true",
                   help="NHWC weights/activations (skips MIOpen transposes)")
    p.add_argument("--cast-mode", choices=["autocast", "pure"],
                   default="autocast",
                   help="bf16 via autocast, or pure bf16 weights with an "
                        "fp32 master-weight Adam (no per-op casts)")
    args = p.parse_args()

    assert torch.cuda.is_available(), "bench.py requires an MI355X GPU"

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    assert world == args.gpus, (
        f"--gpus {args.gpus} but WORLD_SIZE={world}: launch N>1 via torchrun "
        f"(python -m torch.distributed.run --nproc-per-node N bench.py --gpus N)")
    if world > 1:
        # per-rank MIOpen find-db: 8 concurrent processes racing one user
        # db file corrupts/serializes the find phase
        os.environ.setdefault("MIOPEN_USER_DB_PATH",
                              f"/tmp/miopen-rank{local_rank}")
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl", init_method="env://")
    device = torch.device(f"cuda:{local_rank}")
    torch.manual_seed(1234 + rank)

    from esr_amd.models import build_model
    from esr_amd.ops.native import require_ext
    ext = require_ext()  # fail loudly if HIP kernels are missing

    model = build_model("ESRNet", inch=2, basech=args.basech,
                        num_frame=args.seqn,
                        upsampler=args.upsampler).to(device)
    if args.channels_last:
        model = model.to(memory_format=torch.channels_last)

    pure_bf16 = args.cast_mode == "pure" and args.dtype == "bf16"
    if pure_bf16:
        model = model.to(torch.bfloat16)

    # one flat gradient buffer -> a single RCCL all-reduce per step
    params = [prm for prm in model.parameters() if prm.requires_grad]
    total = sum(prm.numel() for prm in params)
    flat_grad = torch.zeros(total, device=device,
                            dtype=torch.bfloat16 if pure_bf16 else torch.float32)
    off = 0
    for prm in params:
        prm.grad = flat_grad[off:off + prm.numel()].view_as(prm)
        off += prm.numel()

    if pure_bf16:
        # fp32 master weights; Adam runs on masters, bf16 working copy is
        # refreshed once per step (one foreach cast instead of per-op casts)
        masters = [prm.detach().float().clone() for prm in params]
        master_flat_grad = torch.zeros(total, device=device)
        off = 0
        for mprm in masters:
            mprm.grad = master_flat_grad[off:off + mprm.numel()].view_as(mprm)
            off += mprm.numel()
        optimizer = torch.optim.Adam(masters, lr=1e-3, weight_decay=1e-4,
                                     amsgrad=True, foreach=True,
                                     capturable=not args.no_graphs)
    else:
        masters = None
        optimizer = torch.optim.Adam(params, lr=1e-3, weight_decay=1e-4,
                                     amsgrad=True, foreach=True,
                                     capturable=not args.no_graphs)

    lr_h = args.lr_h or args.lr_size
    lr_w = args.lr_w or args.lr_size
    lr_res = (lr_h, lr_w)
    hr_res = (lr_h * args.scale, lr_w * args.scale)
    n_windows = args.seql - args.seqn + 1
    pools = make_sequence_pool(2, args.seql, n_windows, args.batch,
                               args.seqn, args.window, lr_res, hr_res,
                               device, ext, seed=100 + rank)
    amp_dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                 "fp32": None}[args.dtype]

    # graph replay reads from fixed addresses: one graph per pre-generated
    # data set (no per-step copies), all sharing one memory pool
    static_frames = [pools[0][0]]
    static_gts = [list(pools[0][1])]

    import contextlib

    def autocast():
        if amp_dtype is None or pure_bf16:
            return contextlib.nullcontext()
        # cache_enabled=False: the autocast weight-cast cache is not
        # graph-capture-safe (casts must be recorded into the graph)
        return torch.autocast("cuda", dtype=amp_dtype, cache_enabled=False)

    def fwd_bwd():
        flat_grad.zero_()
        inner = model.module if hasattr(model, "module") else model
        inner.reset_states()
        frames = static_frames[0]
        x = frames.to(torch.bfloat16) if pure_bf16 else frames
        # shared-encoder BPTT: head/encoder run once per UNIQUE frame, the
        # seqn-windows are contiguous frame-major slices (math-identical to
        # the per-window loop — tests/test_model.py::test_forward_sequence_*)
        with autocast():
            preds = inner.forward_sequence(x, args.seqn)
        loss = 0
        for pred, gt in zip(preds, static_gts[0]):
            loss = loss + F.mse_loss(pred.float(), gt)
        loss.backward()
        return loss

    def comm():
        if world > 1:
            dist.all_reduce(flat_grad)
            flat_grad.div_(world)

    def opt_step():
        if pure_bf16:
            master_flat_grad.copy_(flat_grad)       # one bf16->fp32 cast
            optimizer.step()
            with torch.no_grad():
                torch._foreach_copy_(params, masters)  # refresh bf16 weights
        else:
            optimizer.step()

    def set_data(i):
        frames, gts = pools[i % len(pools)]
        static_frames[0] = frames
        static_gts[0] = list(gts)

    def _warmup_side_stream(body):
        """Warmup initializes MIOpen plans + capturable-Adam state tensors,
        but its real optimizer.step()s run on garbage data — snapshot the
        weights first and roll back (in place) before capture."""
        snap_params = params + (masters if pure_bf16 else [])
        snap = [prm.detach().clone() for prm in snap_params]
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                body()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        with torch.no_grad():
            for prm, sv in zip(snap_params, snap):
                prm.copy_(sv)
            for state in optimizer.state.values():
                for v in state.values():
                    if torch.is_tensor(v):
                        v.zero_()
        torch.cuda.synchronize()

    def _capture(body):
        graphs = []
        pool_handle = None
        for i in range(len(pools)):
            set_data(i)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=pool_handle):
                body()
            pool_handle = g.pool()
            graphs.append(g)
        torch.cuda.synchronize()
        return graphs

    def _rccl_capture_works() -> bool:
        """Trial-capture ONE all-reduce and agree on the verdict across all
        ranks, so every rank picks the same capture tier (a rank-divergent
        tier would deadlock the collectives)."""
        if world == 1:
            return True
        ok = 1
        try:
            probe = torch.ones(8, device=device)
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                dist.all_reduce(probe)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                dist.all_reduce(probe)
            torch.cuda.synchronize()
            g.replay()
            torch.cuda.synchronize()
        except Exception:
            ok = 0
        verdict = torch.tensor([ok], device=device)
        dist.all_reduce(verdict, op=dist.ReduceOp.MIN)
        return bool(verdict.item())

    # Tiered capture: (a) whole step incl. RCCL all-reduce + Adam in the
    # graph; (b) fwd+bwd in the graph, comm + Adam eager; (c) fully eager.
    graphs = None
    graph_mode = "eager"
    if not args.no_graphs:
        comm_in_graph = _rccl_capture_works()

        def full_step():
            fwd_bwd()
            comm()
            opt_step()
        try:
            if not comm_in_graph:
                raise RuntimeError("RCCL graph capture unavailable")
            _warmup_side_stream(full_step)
            graphs = _capture(full_step)
            graph_mode = "full"
        except Exception as e:
            print(f"[bench] full-step capture unavailable ({type(e).__name__});"
                  f" trying fwd+bwd-only capture", flush=True)
            torch.cuda.synchronize()
            try:
                _warmup_side_stream(fwd_bwd)
                graphs = _capture(fwd_bwd)
                graph_mode = "fwd_bwd"
            except Exception as e2:
                print(f"[bench] capture failed ({type(e2).__name__}: {e2}); "
                      f"eager fallback", flush=True)
                graphs = None
        if graph_mode != "full":
            # capturable Adam steps on-device tensors; for eager modes use a
            # plain (faster host-side) Adam
            optimizer = torch.optim.Adam(masters if pure_bf16 else params,
                                         lr=1e-3, weight_decay=1e-4,
                                         amsgrad=True, foreach=True)

    def step(i):
        if graph_mode == "full":
            graphs[i % len(graphs)].replay()
        elif graph_mode == "fwd_bwd":
            graphs[i % len(graphs)].replay()
            comm()
            opt_step()
        else:
            set_data(i)
            fwd_bwd()
            comm()
            opt_step()

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

    metric_name = args.metric_suffix or \
        f"SR event-frames/sec (whole node), {args.scale}x NFS-syn"
    if rank == 0:
        print(json.dumps({
            "metric": metric_name,
            "value": round(value, 2),
            "unit": "frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "peak_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 2),
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
                "input": f"{lr_h}x{lr_w}->{lr_h * args.scale}x{lr_w * args.scale}",
                "scale": args.scale,
                "upsampler": args.upsampler,
                "hip_graphs": graph_mode,
                "parallelism": f"dp{world}",
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
