"""ESIM-style simulator tests."""

import numpy as np
import pytest

from esr_amd.data.simulate import (frames_to_event_store,
                                   sample_contrast_thresholds,
                                   simulate_events)
from esr_amd.data.store import EventStore


def _moving_bar(T=8, H=32, W=32):
    frames = np.zeros((T, H, W))
    for t in range(T):
        x = 4 + t * 3
        frames[t, :, x:x + 4] = 1.0
    ts = np.linspace(0, 0.1, T)
    return frames, ts


def test_simulator_basic_properties():
    frames, ts = _moving_bar()
    ev = simulate_events(frames, ts, cp=0.3, cn=0.3)
    assert ev.shape[0] == 4 and ev.shape[1] > 0
    assert (np.diff(ev[2]) >= 0).all()            # sorted by time
    assert set(np.unique(ev[3])) <= {-1.0, 1.0}
    assert ev[0].min() >= 0 and ev[0].max() < 32
    # a moving bright bar generates both ON (leading) and OFF (trailing)
    assert (ev[3] > 0).any() and (ev[3] < 0).any()


def test_event_count_scales_with_threshold():
    frames, ts = _moving_bar()
    n_lo = simulate_events(frames, ts, cp=0.1, cn=0.1).shape[1]
    n_hi = simulate_events(frames, ts, cp=0.5, cn=0.5).shape[1]
    assert n_lo > n_hi


def test_static_scene_silent():
    frames = np.full((5, 8, 8), 0.5)
    ev = simulate_events(frames, np.linspace(0, 1, 5))
    assert ev.shape[1] == 0


def test_threshold_sampling_distribution():
    rng = np.random.default_rng(0)
    cps, cns = zip(*(sample_contrast_thresholds(rng) for _ in range(200)))
    assert 0.05 <= min(cps) and max(cps) <= 0.5
    assert all(0.01 <= c <= 1.0 for c in cns)


def test_frames_to_event_store(tmp_path):
    frames, ts = _moving_bar(T=6, H=64, W=64)
    p = tmp_path / "sim.evs"
    frames_to_event_store(p, frames, ts, levels=(1, 2, 4), seed=1)
    s = EventStore(p)
    assert set(s.groups) == {"ori", "down2", "down4"}
    assert s.num_events("ori") > s.num_events("down4") > 0
    assert s.num_images == 6
