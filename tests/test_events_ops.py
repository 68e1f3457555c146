"""Event-representation op tests.

Includes the reference's key conversion invariant: random count stack ->
redistribute to events -> re-splat == original
(ESR:dataloader/encodings.py:673-696), plus oracle comparisons against a
direct per-event Python loop.
"""

import torch

from esr_amd.ops import (events_to_channels, events_to_image,
                         events_to_stack_no_polarity, events_to_voxel,
                         redistribute_count, redistribute_stack,
                         stack_to_count, event_formatting, normalize_events,
                         scaled_count_encoding)


def _loop_image(xs, ys, ws, size):
    H, W = size
    img = torch.zeros(H, W)
    for x, y, w in zip(xs.tolist(), ys.tolist(), ws.tolist()):
        xi, yi = int(x), int(y)
        if 0 <= xi < W and 0 <= yi < H:
            img[yi, xi] += w
    return img


def test_events_to_image_matches_loop():
    g = torch.Generator().manual_seed(0)
    n = 500
    xs = torch.rand(n, generator=g) * 12 - 1   # includes out-of-range
    ys = torch.rand(n, generator=g) * 12 - 1
    ws = torch.randn(n, generator=g)
    out = events_to_image(xs.floor(), ys.floor(), ws, (10, 10))
    ref = _loop_image(xs.floor(), ys.floor(), ws, (10, 10))
    assert torch.allclose(out, ref, atol=1e-5)


def test_events_to_channels_counts():
    xs = torch.tensor([0.0, 1.0, 1.0, 2.0])
    ys = torch.tensor([0.0, 1.0, 1.0, 2.0])
    ps = torch.tensor([1.0, -1.0, -1.0, 1.0])
    cnt = events_to_channels(xs, ys, ps, (3, 3))
    assert cnt.shape == (2, 3, 3)
    assert cnt[0, 0, 0] == 1 and cnt[0, 2, 2] == 1
    assert cnt[1, 1, 1] == 2
    assert (cnt >= 0).all()


def test_stack_bins_right_inclusive():
    # 5 events at known times; B=2 bins over [0, 1]
    xs = torch.zeros(5)
    ys = torch.zeros(5)
    ts = torch.tensor([0.0, 0.2, 0.5, 0.7, 1.0])
    ps = torch.ones(5)
    stack = events_to_stack_no_polarity(xs, ys, ts, ps, 2, (2, 2))
    # edge at ~0.5 -> the t=0.5 event is in bin 0 (right-inclusive)
    assert stack[0, 0, 0] == 3 and stack[1, 0, 0] == 2


def test_degenerate_inputs():
    z = torch.zeros(2)
    out = events_to_stack_no_polarity(z, z, z, z, 4, (8, 8))
    assert out.shape == (4, 8, 8) and out.sum() == 0


def test_voxel_temporal_bilinear():
    xs = torch.tensor([1.0])
    ys = torch.tensor([1.0])
    ts = torch.tensor([0.25])   # normalized
    ps = torch.tensor([1.0])
    v = events_to_voxel(xs, ys, ts, ps, 5, (3, 3))
    # tn = 0.25*4 = 1.0 -> all mass in bin 1
    assert torch.isclose(v[1, 1, 1], torch.tensor(1.0))
    ts = torch.tensor([0.30])   # tn = 1.2 -> 0.8 in bin1, 0.2 in bin2
    v = events_to_voxel(xs, ys, ts, ps, 5, (3, 3))
    assert torch.isclose(v[1, 1, 1], torch.tensor(0.8))
    assert torch.isclose(v[2, 1, 1], torch.tensor(0.2))


def test_roundtrip_stack_redistribute():
    """The reference's round-trip invariant
    (ESR:dataloader/encodings.py:673-696)."""
    g = torch.Generator().manual_seed(3)
    stack = torch.randint(-5, 15, (2, 10, 4, 4), generator=g).float()
    cloud = redistribute_stack(stack, mode="linear")
    for b in range(stack.size(0)):
        ev = cloud[b]
        nz = ev.abs().sum(1) > 0
        ev = ev[nz]
        rebuilt = torch.zeros_like(stack[0])
        for x, y, t, p in ev.tolist():
            # right-edge-inclusive binning (timestamps live in
            # (c/C + 1/(100C), (c+1)/C]); 1e-6 margin absorbs fp32 rounding
            c = max(min(int((t - 1e-6) * 10), 9), 0)
            rebuilt[c, int(y), int(x)] += p
        assert torch.allclose(rebuilt, stack[b]), f"batch {b} mismatch"


def test_roundtrip_count_redistribute():
    g = torch.Generator().manual_seed(5)
    cnt = torch.randint(0, 6, (2, 2, 6, 6), generator=g).float()
    cloud = redistribute_count(cnt, mode="random")
    for b in range(cnt.size(0)):
        ev = cloud[b]
        nz = ev.abs().sum(1) > 0
        ev = ev[nz]
        pos = torch.zeros(6, 6)
        neg = torch.zeros(6, 6)
        for x, y, t, p in ev.tolist():
            if p > 0:
                pos[int(y), int(x)] += 1
            else:
                neg[int(y), int(x)] += 1
        assert torch.allclose(pos, cnt[b, 0])
        assert torch.allclose(neg, cnt[b, 1])


def test_redistribute_sorted_and_padded():
    stack = torch.zeros(2, 4, 3, 3)
    stack[0, 1, 1, 1] = 5
    stack[0, 3, 0, 2] = -2
    cloud = redistribute_stack(stack)
    ts = cloud[0, :7, 2]
    assert (ts[1:] >= ts[:-1]).all()
    assert cloud[1].abs().sum() == 0          # empty item zero-padded


def test_stack_to_count():
    stack = torch.tensor([[[[2.0, -1.0], [0.0, 3.0]],
                           [[-2.0, 0.0], [1.0, -1.0]]]])  # [1,2,2,2]
    cnt = stack_to_count(stack)
    assert cnt.shape == (1, 2, 2, 2)
    assert cnt[0, 0, 0, 0] == 2 and cnt[0, 1, 0, 0] == 2
    assert cnt[0, 0, 1, 1] == 3 and cnt[0, 1, 1, 1] == 1


def test_event_formatting_normalizes_t():
    import numpy as np
    ev = np.stack([np.arange(5.0), np.arange(5.0),
                   np.linspace(10.0, 20.0, 5), np.ones(5)])
    out = event_formatting(ev)
    assert out[2].min() == 0
    assert abs(out[2].max().item() - 1.0) < 1e-4


def test_scaled_encoding_resplat():
    ev = torch.tensor([[1.0, 3.0], [1.0, 3.0], [0.1, 0.9], [1.0, -1.0]])
    norm = normalize_events(ev, (4, 4))
    up = scaled_count_encoding(norm, (8, 8), "cnt")
    assert up.shape == (2, 8, 8)
    assert up[0, 2, 2] == 1 and up[1, 6, 6] == 1


def test_voxel_matches_loop_oracle():
    """Temporal-bilinear voxel vs a per-event loop (independent oracle)."""
    g = torch.Generator().manual_seed(11)
    n, B, H, W = 200, 5, 6, 7
    xs = (torch.rand(n, generator=g) * W).floor()
    ys = (torch.rand(n, generator=g) * H).floor()
    ts = torch.rand(n, generator=g)
    ps = torch.randint(0, 2, (n,), generator=g).float() * 2 - 1
    v = events_to_voxel(xs, ys, ts, ps, B, (H, W))
    ref = torch.zeros(B, H, W)
    for x, y, t, p in zip(xs.tolist(), ys.tolist(), ts.tolist(), ps.tolist()):
        for b in range(B):
            w = max(0.0, 1.0 - abs(t * (B - 1) - b))
            ref[b, int(y), int(x)] += p * w
    assert torch.allclose(v, ref, atol=1e-4), (v - ref).abs().max().item()


def test_redistribute_polarity_stack_5d():
    g = torch.Generator().manual_seed(12)
    stack = torch.randint(0, 4, (2, 2, 3, 4, 4), generator=g).float()
    stack[:, 1] *= -1  # negative polarity plane holds negative values
    cloud = redistribute_stack(stack, mode="linear")
    for b in range(2):
        ev = cloud[b]
        ev = ev[ev.abs().sum(1) > 0]
        assert ev.shape[0] == int(stack[b].abs().sum().item())
        # channel-0 cells (positive values) emit +1, channel-1 emit -1
        n_pos = int(stack[b, 0].abs().sum().item())
        assert (ev[:, 3] > 0).sum().item() == n_pos


def test_scaled_encoding_stack_and_events_modes():
    # >3 events (the reference's degenerate early-out returns zeros below)
    ev = torch.tensor([[1.0, 1.0, 3.0, 3.0, 2.0],
                       [1.0, 1.0, 3.0, 3.0, 2.0],
                       [0.1, 0.2, 0.8, 0.9, 0.5],
                       [1.0, 1.0, -1.0, -1.0, 1.0]])
    norm = normalize_events(ev, (4, 4))
    st = scaled_count_encoding(norm, (8, 8), "stack", time_bins=2)
    assert st.shape == (2, 8, 8)
    assert st[0, 2, 2] == 2 and st[1, 6, 6] == -2
    evm = scaled_count_encoding(norm, (8, 8), "events")
    assert evm.shape == (4, 5)
    assert evm[0, 0] == 2 and evm[1, 2] == 6


def test_redistribute_capacity_cpu():
    """The torch path honours an explicit capacity like the HIP pipeline:
    fixed output shape, zero padding, truncation when over."""
    from esr_amd.ops.events import redistribute_stack
    g = torch.Generator().manual_seed(4)
    stack = torch.randint(0, 3, (2, 2, 8, 8), generator=g).float()
    n0 = int(stack[0].abs().sum())
    full = redistribute_stack(stack, mode="linear")
    padded = redistribute_stack(stack, mode="linear", capacity=n0 + 16)
    assert padded.shape[1] == n0 + 16
    assert torch.equal(padded[:, : full.shape[1]], full)
    assert (padded[:, full.shape[1]:] == 0).all()
    small = redistribute_stack(stack, mode="linear", capacity=5)
    assert small.shape[1] == 5
    assert torch.equal(small[0], full[0, :5])
