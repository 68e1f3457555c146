#!/usr/bin/env python3
"""Generate synthetic EVS datasets (emitter-based or ESIM-simulated).

Replaces the reference's offline NFS->ESIM pipeline
(ESR:generate_dataset/syn_nfs_rgb.py) for this network-free environment.

  python tools/make_synth_data.py --out data/synth --sequences 4 \
      --resolution 256 --events 200000 --split 0.75
  python tools/make_synth_data.py --out data/sim --mode esim --frames 32
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", required=True)
    p.add_argument("--sequences", type=int, default=4)
    p.add_argument("--resolution", type=int, default=256)
    p.add_argument("--events", type=int, default=200_000)
    p.add_argument("--frames", type=int, default=32, help="esim mode")
    p.add_argument("--mode", choices=["emitter", "esim"], default="emitter")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--split", type=float, default=0.75,
                   help="train fraction; writes train/valid datalists")
    args = p.parse_args()

    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    paths = []
    if args.mode == "emitter":
        from esr_amd.data.synthetic import write_synthetic_store
        for i in range(args.sequences):
            pth = out / f"seq{i:03d}.evs"
            write_synthetic_store(pth, (args.resolution, args.resolution),
                                  args.events, seed=args.seed + i)
            paths.append(str(pth))
    else:
        import numpy as np
        from esr_amd.data.simulate import frames_to_event_store
        rng = np.random.default_rng(args.seed)
        for i in range(args.sequences):
            # smooth random-texture pan (NFS-like camera motion)
            H = W = args.resolution
            tex = rng.random((H * 2, W * 2))
            import scipy.ndimage as ndi
            tex = ndi.gaussian_filter(tex, 4)
            tex = (tex - tex.min()) / (np.ptp(tex) + 1e-9)
            frames = []
            for t in range(args.frames):
                dy = int(t * H / (2 * args.frames))
                dx = int(t * W / (2 * args.frames))
                frames.append(tex[dy:dy + H, dx:dx + W])
            frames = np.stack(frames)
            ts = np.linspace(0, 0.5, args.frames)
            pth = out / f"sim{i:03d}.evs"
            frames_to_event_store(pth, frames, ts, seed=args.seed + i)
            paths.append(str(pth))

    n_train = max(int(len(paths) * args.split), 1)
    with open(out / "train_datalist.txt", "w") as f:
        f.write("\n".join(paths[:n_train]) + "\n")
    with open(out / "valid_datalist.txt", "w") as f:
        f.write("\n".join(paths[n_train:] or paths[-1:]) + "\n")
    print(f"wrote {len(paths)} sequences under {out}")


if __name__ == "__main__":
    main()
