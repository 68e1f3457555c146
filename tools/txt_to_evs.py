#!/usr/bin/env python3
"""Convert a text event file to an EVS store with derived down-scale groups.

Text format (parity: ESR:generate_dataset/tools/txt_to_h5.py): one event
per line, `t x y p` (whitespace-separated; p in {0,1} or {-1,1}).

  python tools/txt_to_evs.py events.txt out.evs --height 480 --width 640
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from esr_amd.data.store import EventStoreWriter  # noqa: E402

NAMES = {1: "ori", 2: "down2", 4: "down4", 8: "down8", 16: "down16"}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("txt_path")
    p.add_argument("out_path")
    p.add_argument("--height", type=int, required=True)
    p.add_argument("--width", type=int, required=True)
    p.add_argument("--levels", type=int, nargs="+", default=[1, 2, 4, 8, 16])
    p.add_argument("--skip-header", type=int, default=0)
    args = p.parse_args()

    data = np.loadtxt(args.txt_path, skiprows=args.skip_header)
    ts, xs, ys, ps = data[:, 0], data[:, 1], data[:, 2], data[:, 3]
    ps = np.where(ps > 0, 1.0, -1.0)
    order = np.argsort(ts, kind="stable")
    ts, xs, ys, ps = ts[order], xs[order], ys[order], ps[order]

    with EventStoreWriter(args.out_path, (args.height, args.width)) as w:
        for lvl in args.levels:
            sub = slice(None, None, lvl * lvl)   # 1/k^2 count thinning
            w.add_group(NAMES[lvl], np.floor(xs[sub] / lvl),
                        np.floor(ys[sub] / lvl), ts[sub], ps[sub])
    print(f"wrote {args.out_path}: {len(ts)} events")


if __name__ == "__main__":
    main()
