#!/usr/bin/env python3
"""Render an EVS sequence's windows to PNG count maps / stack grids
(parity: ESR:myutils/event_visual_example.py — the visualization example CLI).

  python tools/visualize_events.py path/to/seq.evs --out viz --windows 8
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from esr_amd.data.dataset import EventSRDataset  # noqa: E402
from esr_amd.utils.vis import EventVisualizer, plot_event_stack  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("seq_path")
    p.add_argument("--out", default="viz")
    p.add_argument("--windows", type=int, default=8)
    p.add_argument("--ori_scale", default="down4")
    p.add_argument("--scale", type=int, default=2)
    p.add_argument("--window", type=int, default=2048)
    p.add_argument("--sliding_window", type=int, default=1024)
    p.add_argument("--time_bins", type=int, default=4)
    args = p.parse_args()

    cfg = {"scale": args.scale, "ori_scale": args.ori_scale,
           "time_bins": args.time_bins, "need_gt_frame": False,
           "need_gt_events": True, "mode": "events",
           "window": args.window, "sliding_window": args.sliding_window,
           "data_augment": {"enabled": False, "augment": [],
                            "augment_prob": []},
           "hot_filter": {"enabled": False},
           "sequence": {"sequence_length": 1, "seqn": 1, "step_size": None,
                        "pause": {"enabled": False,
                                  "proba_pause_when_running": 0,
                                  "proba_pause_when_paused": 0}}}
    ds = EventSRDataset(args.seq_path, cfg)
    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    vis = EventVisualizer()
    n = min(args.windows, len(ds))
    for i in range(n):
        item = ds.__getitem__(i, seed=0)
        hwc = lambda t: t.numpy().transpose(1, 2, 0)  # noqa: E731
        vis.plot_event_cnt(hwc(item["inp_cnt"]), True,
                           str(out / f"{i:04d}_lr_cnt.png"))
        vis.plot_event_cnt(hwc(item["inp_scaled_cnt"]), True,
                           str(out / f"{i:04d}_hr_scaled_cnt.png"))
        vis.plot_event_cnt(hwc(item["gt_cnt"]), True,
                           str(out / f"{i:04d}_gt_cnt.png"))
        plot_event_stack(item["inp_stack"].numpy(), True,
                         str(out / f"{i:04d}_stack.png"))
    print(f"wrote {4 * n} renders under {out}")


if __name__ == "__main__":
    main()
