"""Synthetic multi-scale event-stream generation.

The reference builds its NFS-syn dataset offline with the ESIM simulator at
five spatial scales (ESR:generate_dataset/syn_nfs_rgb.py:66-133).  There is
no network or ESIM here, so benchmarks and tests use this generator: events
are emitted by moving Gaussian "emitters" (temporally coherent, spatially
clustered — the statistics that matter to the splatting ops), written at
the finest resolution, and each down{k} group is derived from the ori
stream by coordinate scaling + 1/k^2 count thinning, which preserves the
reference's GT-alignment invariant (a time range holds ~scale^2 more GT
events than input events, ESR:dataloader/h5dataset.py:451-475).
"""

from __future__ import annotations

from pathlib import Path

import numpy as np

from .store import EventStoreWriter

__all__ = ["generate_events", "write_synthetic_store", "make_synthetic_dataset"]


def generate_events(num_events: int, resolution, seed: int = 0,
                    num_blobs: int = 12, duration: float = 1.0) -> np.ndarray:
    """Return [4, N] (x, y, t, p) events from moving emitters."""
    H, W = resolution
    rng = np.random.default_rng(seed)
    t = np.sort(rng.random(num_events)) * duration
    blob = rng.integers(0, num_blobs, num_events)

    x0 = rng.random(num_blobs) * W
    y0 = rng.random(num_blobs) * H
    vx = (rng.random(num_blobs) - 0.5) * W
    vy = (rng.random(num_blobs) - 0.5) * H
    radius = (0.02 + 0.08 * rng.random(num_blobs)) * max(H, W)
    pol_bias = rng.random(num_blobs)

    xs = np.mod(x0[blob] + vx[blob] * t / duration +
                rng.normal(0, radius[blob]), W)
    ys = np.mod(y0[blob] + vy[blob] * t / duration +
                rng.normal(0, radius[blob]), H)
    ps = np.where(rng.random(num_events) < pol_bias[blob], 1.0, -1.0)
    return np.stack([np.floor(xs), np.floor(ys), t, ps])


def _render_image(xs, ys, resolution) -> np.ndarray:
    H, W = resolution
    img = np.zeros((H, W), dtype=np.float64)
    np.add.at(img, (ys.astype(int).clip(0, H - 1),
                    xs.astype(int).clip(0, W - 1)), 1.0)
    if img.max() > 0:
        img = img / img.max()
    return (img * 255).astype(np.uint8)


def write_synthetic_store(path, resolution=(256, 256), num_events=200_000,
                          levels=(1, 2, 4, 8, 16), num_images=8, seed=0):
    """Create one synthetic EVS sequence with all down-scale groups."""
    H, W = resolution
    ev = generate_events(num_events, resolution, seed=seed)
    names = {1: "ori", 2: "down2", 4: "down4", 8: "down8", 16: "down16"}
    with EventStoreWriter(path, resolution) as w:
        for lvl in levels:
            sub = ev[:, :: lvl * lvl]          # 1/k^2 count thinning
            w.add_group(names[lvl],
                        np.floor(sub[0] / lvl), np.floor(sub[1] / lvl),
                        sub[2], sub[3])
        if num_images:
            step = num_events // num_images
            imgs, ts = [], []
            for i in range(num_images):
                sl = slice(i * step, (i + 1) * step)
                imgs.append(_render_image(ev[0, sl], ev[1, sl], resolution))
                ts.append(float(ev[2, sl].mean()))
            w.add_images(np.stack(imgs), ts)
    return str(path)


def make_synthetic_dataset(root, num_sequences=2, resolution=(256, 256),
                           num_events=200_000, seed=0, datalist_name="datalist.txt"):
    """Write N sequences + a datalist txt; returns the datalist path."""
    root = Path(root)
    root.mkdir(parents=True, exist_ok=True)
    paths = []
    for i in range(num_sequences):
        p = root / f"seq{i:03d}.evs"
        write_synthetic_store(p, resolution, num_events, seed=seed + i)
        paths.append(str(p))
    datalist = root / datalist_name
    with open(datalist, "w") as f:
        f.write("\n".join(paths) + "\n")
    return str(datalist)
