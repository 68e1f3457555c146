#!/usr/bin/env python3
"""Convert a reference-format HDF5 sequence to an EVS store.

The reference's HDF5 schema (ESR:generate_dataset/tools/event_packagers.py:
119-224): per-scale groups '{ori,down2,down4,down8,down16}_events' with
xs/ys/ts/ps datasets, an 'ori_images' group of frames with 'timestamp'
attrs, and a 'sensor_resolution' file attribute.  Requires h5py (not part
of this image; the tool degrades with a clear error).

  python tools/convert_h5.py input.h5 output.evs
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def convert(h5_path: str, out_path: str) -> str:
    try:
        import h5py
    except ImportError as e:
        raise SystemExit("h5py is not installed in this environment; "
                         "run this converter where h5py is available") from e
    import numpy as np
    from esr_amd.data.store import EventStoreWriter, GROUP_LEVELS

    with h5py.File(h5_path, "r") as f:
        res = list(f.attrs["sensor_resolution"])
        with EventStoreWriter(out_path, res) as w:
            for prefix in GROUP_LEVELS:
                grp = f"{prefix}_events"
                if grp in f:
                    g = f[grp]
                    w.add_group(prefix, g["xs"][:], g["ys"][:],
                                g["ts"][:], g["ps"][:])
            if "ori_images" in f:
                imgs, ts = [], []
                for name in sorted(f["ori_images"]):
                    d = f[f"ori_images/{name}"]
                    imgs.append(d[:])
                    ts.append(float(d.attrs["timestamp"]))
                if imgs:
                    w.add_images(np.stack(imgs), ts)
    return out_path


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("h5_path")
    p.add_argument("out_path")
    args = p.parse_args()
    print(convert(args.h5_path, args.out_path))
