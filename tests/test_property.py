"""Property-based tests (hypothesis) for the event-op invariants."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from esr_amd.ops import (events_to_channels, events_to_stack_no_polarity,
                         redistribute_stack, stack_to_count)


@st.composite
def event_batches(draw):
    n = draw(st.integers(min_value=4, max_value=200))
    H = draw(st.integers(min_value=2, max_value=16))
    W = draw(st.integers(min_value=2, max_value=16))
    seed = draw(st.integers(min_value=0, max_value=2 ** 16))
    g = torch.Generator().manual_seed(seed)
    xs = (torch.rand(n, generator=g) * (W + 2) - 1).floor()
    ys = (torch.rand(n, generator=g) * (H + 2) - 1).floor()
    ts = torch.sort(torch.rand(n, generator=g)).values
    ps = torch.randint(0, 2, (n,), generator=g).float() * 2 - 1
    return xs, ys, ts, ps, (H, W)


@settings(max_examples=40, deadline=None)
@given(event_batches())
def test_count_conservation(batch):
    """Total counts == number of in-range events; channels nonnegative."""
    xs, ys, ts, ps, size = batch
    H, W = size
    cnt = events_to_channels(xs, ys, ps, size)
    in_range = ((xs >= 0) & (xs < W) & (ys >= 0) & (ys < H)).sum().item()
    assert cnt.sum().item() == in_range
    assert (cnt >= 0).all()


@settings(max_examples=40, deadline=None)
@given(event_batches(), st.integers(min_value=1, max_value=6))
def test_stack_sums_to_signed_count(batch, B):
    """Summing a stack over time bins equals the signed per-pixel total."""
    xs, ys, ts, ps, size = batch
    stack = events_to_stack_no_polarity(xs, ys, ts, ps, B, size)
    cnt = events_to_channels(xs, ys, ps, size)
    signed = cnt[0] - cnt[1]
    assert torch.allclose(stack.sum(0), signed, atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(st.integers(min_value=0, max_value=2 ** 16),
       st.integers(min_value=1, max_value=6),
       st.integers(min_value=2, max_value=6))
def test_redistribute_count_preserving(seed, C, HW):
    """redistribute emits exactly |v| events per cell with sign(v)."""
    g = torch.Generator().manual_seed(seed)
    stack = torch.randint(-4, 7, (1, C, HW, HW), generator=g).float()
    cloud = redistribute_stack(stack, mode="linear")
    ev = cloud[0]
    ev = ev[ev.abs().sum(1) > 0]
    assert ev.shape[0] == int(stack.abs().sum().item())
    # per-pixel signed totals survive the round trip
    rebuilt_cnt = stack_to_count(stack)
    pos = torch.zeros(HW, HW)
    neg = torch.zeros(HW, HW)
    for x, y, t, p in ev.tolist():
        if p > 0:
            pos[int(y), int(x)] += 1
        else:
            neg[int(y), int(x)] += 1
    assert torch.allclose(pos, rebuilt_cnt[0, 0])
    assert torch.allclose(neg, rebuilt_cnt[0, 1])


def test_multiworker_loader_pickling(synth_datalist):
    """num_workers>0 exercises collate-class pickling in worker procs."""
    from esr_amd.data import SequenceDataLoader
    ds_cfg = {
        "scale": 2, "ori_scale": "down4", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 1024, "sliding_window": 512,
        "data_augment": {"enabled": False, "augment": [], "augment_prob": []},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 4, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0,
                               "proba_pause_when_paused": 0}},
    }
    for collate in (None, "shared"):
        cfg = {"use_ddp": False, "path_to_datalist_txt": synth_datalist,
               "batch_size": 2, "shuffle": False, "num_workers": 2,
               "pin_memory": False, "drop_last": True, "dataset": ds_cfg}
        if collate:
            cfg["collate"] = collate
        loader = SequenceDataLoader(cfg)
        batch = next(iter(loader))
        if collate == "shared":
            assert batch["frames"].shape[1] == 4
        else:
            assert len(batch) == 2
        del loader
