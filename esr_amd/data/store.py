"""EVS — the framework's on-disk event store.

The reference keeps each sequence in one HDF5 file with per-scale event
groups and an image group (schema:
ESR:generate_dataset/tools/event_packagers.py:119-224).  h5py is not part of
this environment, and HDF5 chunk decompression in dataloader workers is pure
CPU overhead for data this shape — an EVS sequence is a *directory* of raw
little-endian .npy arrays (memory-mapped on read; the OS page cache does the
work) plus a JSON metadata file:

    seq.evs/
      meta.json                  {"sensor_resolution": [H, W],
                                  "groups": {"ori": N, "down2": N2, ...},
                                  "num_images": M}
      <group>_xs.npy  (uint16)   event x coordinates
      <group>_ys.npy  (uint16)   event y coordinates
      <group>_ts.npy  (float64)  event timestamps (seconds, sorted)
      <group>_ps.npy  (int8)     event polarities in {-1, +1}
      images.npy      (uint8)    [M, H, W] frames (optional)
      image_ts.npy    (float64)  [M] frame timestamps (optional)

Groups follow the reference's naming: 'ori', 'down2', 'down4', 'down8',
'down16' (and 'down8_real' for real-sensor captures).
"""

from __future__ import annotations

import json
import os
from pathlib import Path

import numpy as np

__all__ = ["EventStoreWriter", "EventStore", "GROUP_LEVELS"]

GROUP_LEVELS = {"ori": 1, "down2": 2, "down4": 4, "down8": 8, "down16": 16,
                "down8_real": 8}

_FIELDS = (("xs", np.uint16), ("ys", np.uint16), ("ts", np.float64), ("ps", np.int8))


class EventStoreWriter:
    """Creates an EVS sequence directory."""

    def __init__(self, path, sensor_resolution):
        self.path = Path(path)
        self.path.mkdir(parents=True, exist_ok=True)
        self.meta = {"sensor_resolution": [int(sensor_resolution[0]),
                                           int(sensor_resolution[1])],
                     "groups": {}, "num_images": 0}

    def add_group(self, prefix: str, xs, ys, ts, ps):
        n = len(ts)
        assert len(xs) == len(ys) == len(ps) == n
        for (name, dt), arr in zip(_FIELDS, (xs, ys, ts, ps)):
            np.save(self.path / f"{prefix}_{name}.npy",
                    np.asarray(arr).astype(dt, copy=False))
        self.meta["groups"][prefix] = int(n)

    def add_images(self, images, timestamps):
        images = np.asarray(images, dtype=np.uint8)
        np.save(self.path / "images.npy", images)
        np.save(self.path / "image_ts.npy",
                np.asarray(timestamps, dtype=np.float64))
        self.meta["num_images"] = int(images.shape[0])

    def close(self):
        with open(self.path / "meta.json", "w") as f:
            json.dump(self.meta, f)

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class EventStore:
    """Read-only, memory-mapped view of an EVS sequence directory."""

    def __init__(self, path):
        self.path = Path(path)
        with open(self.path / "meta.json") as f:
            self.meta = json.load(f)
        self.sensor_resolution = list(self.meta["sensor_resolution"])
        self._mm: dict[str, np.ndarray] = {}

    @property
    def groups(self):
        return dict(self.meta["groups"])

    def _arr(self, name) -> np.ndarray:
        if name not in self._mm:
            self._mm[name] = np.load(self.path / f"{name}.npy", mmap_mode="r")
        return self._mm[name]

    def num_events(self, prefix: str) -> int:
        return int(self.meta["groups"][prefix])

    def ts(self, prefix: str) -> np.ndarray:
        return self._arr(f"{prefix}_ts")

    def events(self, prefix: str, idx0: int, idx1: int) -> np.ndarray:
        """Return a [4, n] float64 array (x, y, t, p) — the layout the
        event-formatting op expects (ESR:dataloader/h5dataset.py:492-498)."""
        sl = slice(idx0, idx1)
        xs = np.asarray(self._arr(f"{prefix}_xs")[sl], dtype=np.float64)
        ys = np.asarray(self._arr(f"{prefix}_ys")[sl], dtype=np.float64)
        ts = np.asarray(self._arr(f"{prefix}_ts")[sl], dtype=np.float64)
        ps = np.asarray(self._arr(f"{prefix}_ps")[sl], dtype=np.float64)
        return np.stack([xs, ys, ts, ps])

    @property
    def num_images(self) -> int:
        return int(self.meta.get("num_images", 0))

    def image(self, i: int) -> np.ndarray:
        return np.asarray(self._arr("images")[i])

    def image_ts(self) -> np.ndarray:
        return np.asarray(self._arr("image_ts"))


def is_event_store(path) -> bool:
    return os.path.isfile(os.path.join(path, "meta.json"))
