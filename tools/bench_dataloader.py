#!/usr/bin/env python3
"""Measure the CPU dataloader feed rate vs worker count.

8-GPU readiness: a node feeding 8 ranks at the flagship step rate needs
~23k items/s aggregate (batch 64 x 8 ranks x ~7.5 steps/s x 6 windows
-> items = sequence elements; see profiles/README.md).  This measures
items/s of SequenceDataLoader at several num_workers on the actual host
and reports the worker count needed per rank.

Usage: python tools/bench_dataloader.py [--seconds 12] [--workers 0 2 4 8]
"""

import argparse
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def _dl_config(datalist, workers, batch):
    ds = {
        "scale": 2, "ori_scale": "down2", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 2048, "sliding_window": 1024,
        "data_augment": {"enabled": True,
                         "augment": ["Horizontal", "Vertical", "Polarity"],
                         "augment_prob": [0.5, 0.5, 0.5]},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 8, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0.05,
                               "proba_pause_when_paused": 0.9}},
    }
    return {"use_ddp": False, "path_to_datalist_txt": str(datalist),
            "batch_size": batch, "shuffle": True, "num_workers": workers,
            "pin_memory": False, "drop_last": True, "dataset": ds,
            "persistent_workers": workers > 0}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=12.0)
    ap.add_argument("--workers", type=int, nargs="*", default=[0, 2, 4, 8])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--fields", action="store_true",
                    help="restrict items to the trainer's 3 fields")
    args = ap.parse_args()

    from esr_amd.data import SequenceDataLoader, make_synthetic_dataset

    tmp = tempfile.mkdtemp(prefix="dlbench")
    datalist = make_synthetic_dataset(Path(tmp), num_sequences=4,
                                      resolution=(128, 128),
                                      num_events=400_000, seed=3)
    print(f"{'workers':>8s} {'items/s':>10s} {'seq items/s':>12s}")
    results = {}
    for w in args.workers:
        cfg = _dl_config(datalist, w, args.batch)
        if args.fields:
            cfg["dataset"]["fields"] = ["inp_scaled_cnt", "gt_cnt", "inp_cnt"]
        dl = SequenceDataLoader(cfg)
        it = iter(dl)
        next(it)  # warm workers
        n = 0
        t0 = time.perf_counter()
        while time.perf_counter() - t0 < args.seconds:
            try:
                next(it)
            except StopIteration:
                it = iter(dl)
                continue
            n += 1
        dt = time.perf_counter() - t0
        # one loader batch = batch sequences x 8 windows = items
        items = n * args.batch * 8
        results[w] = items / dt
        print(f"{w:8d} {items / dt:10.0f} {n * args.batch / dt:12.1f}")
    best = max(results.values())
    need = 23000
    print(f"\nnode target ~{need} items/s over 8 ranks -> "
          f"{need / 8:.0f}/rank; best measured {best:.0f} items/s "
          f"({'OK with' if best >= need / 8 else 'needs more than'} "
          f"measured workers per rank)")


if __name__ == "__main__":
    main()
