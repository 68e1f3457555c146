"""ESIM-style event simulation from frame sequences.

The reference generates its training data offline with the C++ ESIM
simulator over 5 spatial scales with randomized contrast thresholds
(ESR:generate_dataset/syn_nfs_rgb.py:66-133).  This is a vectorized
re-implementation of the same generative model:

  * per pixel, events fire whenever the log-intensity crosses a multiple
    of the contrast threshold (Cp positive / Cn negative), with linear
    interpolation of the crossing time between frames;
  * Cp, Cn are sampled per sequence from the reference's distribution
    (Cp ~ U[0.05, 0.5], Cn ~ N(Cp, 0.03) clipped — ESR:syn_nfs_rgb.py:114-121);
  * the multi-scale groups (ori..down16) are simulated from bicubic-scaled
    copies of the frames like the reference's per-scale ESIM runs, then
    packaged into one EVS store (schema parity:
    ESR:generate_dataset/tools/event_packagers.py:119-224).
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

from .store import EventStoreWriter

__all__ = ["simulate_events", "sample_contrast_thresholds",
           "frames_to_event_store"]

_LOG_EPS = 1e-3


def sample_contrast_thresholds(rng: np.random.Generator):
    """Cp ~ U[0.05, 0.5]; Cn ~ N(Cp, 0.03), clipped to [0.01, 1]
    (parity: ESR:generate_dataset/syn_nfs_rgb.py:114-121)."""
    cp = rng.uniform(0.05, 0.5)
    cn = float(np.clip(rng.normal(cp, 0.03), 0.01, 1.0))
    return cp, cn


def simulate_events(frames: np.ndarray, timestamps: np.ndarray,
                    cp: float = 0.2, cn: float = 0.2,
                    refractory: float = 0.0) -> np.ndarray:
    """Simulate an event stream from a frame sequence.

    frames: [T, H, W] in [0, 1]; timestamps: [T] seconds (increasing).
    Returns [4, N] float64 (x, y, t, p) sorted by t.
    """
    T, H, W = frames.shape
    logf = np.log(frames.astype(np.float64) + _LOG_EPS)
    ref = logf[0].copy()          # per-pixel reference level at last event
    xs_all, ys_all, ts_all, ps_all = [], [], [], []

    yy, xx = np.mgrid[0:H, 0:W]
    for k in range(1, T):
        l0, l1 = logf[k - 1], logf[k]
        t0, t1 = timestamps[k - 1], timestamps[k]
        dl = l1 - ref
        # number of threshold crossings this inter-frame interval
        n_pos = np.floor(np.maximum(dl, 0) / cp).astype(np.int64)
        n_neg = np.floor(np.maximum(-dl, 0) / cn).astype(np.int64)
        for n_cross, C, pol in ((n_pos, cp, 1.0), (n_neg, cn, -1.0)):
            mx = int(n_cross.max()) if n_cross.size else 0
            for i in range(1, mx + 1):
                m = n_cross >= i
                if not m.any():
                    break
                # linear interpolation of the crossing time inside [t0, t1]
                target = ref[m] + pol * C * i
                denom = (l1 - l0)[m]
                frac = np.where(np.abs(denom) > 1e-12,
                                (target - l0[m]) / denom, 0.5)
                frac = np.clip(frac, 0.0, 1.0)
                t = t0 + frac * (t1 - t0)
                xs_all.append(xx[m])
                ys_all.append(yy[m])
                ts_all.append(t)
                ps_all.append(np.full(m.sum(), pol))
        # update reference level to the last crossed threshold
        ref = ref + n_pos * cp - n_neg * cn

    if not ts_all:
        return np.zeros((4, 0))
    xs = np.concatenate(xs_all).astype(np.float64)
    ys = np.concatenate(ys_all).astype(np.float64)
    ts = np.concatenate(ts_all)
    ps = np.concatenate(ps_all)
    order = np.argsort(ts, kind="stable")
    return np.stack([xs[order], ys[order], ts[order], ps[order]])


def _scale_frames(frames: np.ndarray, level: int) -> np.ndarray:
    if level == 1:
        return frames
    t = torch.from_numpy(frames).float().unsqueeze(1)
    H, W = frames.shape[1:]
    out = F.interpolate(t, size=(H // level, W // level), mode="bicubic",
                        align_corners=False).clamp(0, 1)
    return out.squeeze(1).numpy()


def frames_to_event_store(path, frames: np.ndarray, timestamps,
                          levels=(1, 2, 4, 8, 16), cp=None, cn=None,
                          seed: int = 0, store_images: bool = True) -> str:
    """Simulate per-scale event streams from frames and write an EVS store
    (the reference's per-scale ESIM + packager pipeline,
    ESR:generate_dataset/syn_nfs_rgb.py:108-133)."""
    rng = np.random.default_rng(seed)
    if cp is None or cn is None:
        cp, cn = sample_contrast_thresholds(rng)
    names = {1: "ori", 2: "down2", 4: "down4", 8: "down8", 16: "down16"}
    timestamps = np.asarray(timestamps, dtype=np.float64)
    with EventStoreWriter(path, frames.shape[1:]) as w:
        for lvl in levels:
            fr = _scale_frames(frames, lvl)
            ev = simulate_events(fr, timestamps, cp, cn)
            w.add_group(names[lvl], ev[0], ev[1], ev[2], ev[3])
        if store_images:
            w.add_images((frames * 255).astype(np.uint8), timestamps)
    return str(path)
