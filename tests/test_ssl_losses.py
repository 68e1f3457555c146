"""Self-supervised loss stack tests (IWE, contrast-max, photometric)."""

import torch

from esr_amd.loss.flow import AveragedIWE, EventWarping
from esr_amd.loss.reconstruction import BrightnessConstancy
from esr_amd.utils.iwe import compute_pol_iwe, deblur_events, get_interpolation, \
    interpolate
from esr_amd.utils.gradients import Sobel


def _events(B=2, N=64, H=16, W=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    ev = torch.zeros(B, N, 4)
    ev[:, :, 0] = torch.sort(torch.rand(B, N, generator=g), dim=1).values  # ts
    ev[:, :, 1] = (torch.rand(B, N, generator=g) * H).floor()              # y
    ev[:, :, 2] = (torch.rand(B, N, generator=g) * W).floor()              # x
    ev[:, :, 3] = torch.randint(0, 2, (B, N), generator=g) * 2 - 1         # p
    pol = torch.zeros(B, N, 2)
    pol[:, :, 0] = (ev[:, :, 3] > 0).float()
    pol[:, :, 1] = (ev[:, :, 3] < 0).float()
    return ev, pol


def test_zero_flow_iwe_equals_splat():
    ev, pol = _events()
    flow = torch.zeros(2, 2, 16, 16)
    iwe = compute_pol_iwe(flow, ev, (16, 16), pol[:, :, 0:1], pol[:, :, 1:2],
                          flow_scaling=16, round_idx=True)
    # zero flow, round_idx: every event lands on its own pixel
    assert iwe.shape == (2, 2, 16, 16)
    assert iwe.sum() == ev.shape[0] * ev.shape[1]
    from esr_amd.ops import events_to_channels
    ref = events_to_channels(ev[0, :, 2], ev[0, :, 1], ev[0, :, 3], (16, 16))
    assert torch.allclose(iwe[0], ref)


def test_interpolation_weights_sum():
    ev, _ = _events()
    flow = torch.rand(2, ev.shape[1], 2) * 0.01
    idx, w = get_interpolation(ev, flow, 1, (16, 16), 16, round_idx=False)
    img = interpolate(idx.long(), w, (16, 16))
    # bilinear weights of in-range events sum to ~1 per event
    assert img.sum().item() <= ev.shape[0] * ev.shape[1] + 1e-3


def test_event_warping_loss_prefers_compensating_flow():
    # a few dense trajectories moving +3 px in x over the window; the
    # compensating flow must reduce the contrast-max loss
    B, N, H, W = 1, 512, 16, 16
    g = torch.Generator().manual_seed(1)
    ts = torch.sort(torch.rand(B, N, generator=g), dim=1).values
    traj = torch.randint(0, 4, (B, N), generator=g)
    y = (2.0 + traj * 3).float()
    x0 = (2.0 + traj * 2).float()
    x = (x0 + ts * 3).floor().clamp(0, W - 1)
    ev = torch.stack([ts, y, x, torch.ones(B, N)], dim=2)
    pol = torch.cat([torch.ones(B, N, 1), torch.zeros(B, N, 1)], dim=2)

    crit = EventWarping(flow_regul_weight=0.0)
    zero_flow = [torch.zeros(B, 2, H, W)]
    good_flow = [torch.full((B, 2, H, W), 0.0)]
    good_flow[0][:, 0] = 3.0 / max(H, W)   # x-flow compensating the motion
    loss_zero = crit(zero_flow, ev, pol, (H, W))
    loss_good = crit(good_flow, ev, pol, (H, W))
    assert loss_good < loss_zero


def test_averaged_iwe_zero_flow_counts():
    ev, pol = _events(B=1, N=32)
    avg = AveragedIWE((16, 16))
    out = avg(torch.zeros(1, 2, 16, 16), ev, pol)
    assert out.shape == (1, 2, 16, 16)
    # with zero flow each receiving pixel has exactly 1 contributing source
    # pixel (itself), so average == raw count
    from esr_amd.ops import events_to_channels
    ref = events_to_channels(ev[0, :, 2], ev[0, :, 1], ev[0, :, 3], (16, 16))
    assert torch.allclose(out[0], ref)


def test_brightness_constancy_components():
    ev, pol = _events(B=1, N=64)
    bc = BrightnessConstancy((16, 16))
    flow = torch.rand(1, 2, 16, 16) * 0.01
    img = torch.rand(1, 1, 16, 16)
    prev = torch.rand(1, 1, 16, 16)
    cnt = torch.rand(1, 2, 16, 16)
    g = bc.generative_model(flow, img, cnt, ev, pol)
    t = bc.temporal_consistency(flow, prev, img)
    r = bc.regularization(img)
    for v in (g, t, r):
        assert torch.isfinite(v) and v >= 0


def test_sobel_shapes():
    s = Sobel()
    gx, gy = s(torch.rand(2, 1, 8, 8))
    assert gx.shape == (2, 1, 8, 8) and gy.shape == (2, 1, 8, 8)
    # constant image -> zero gradients
    gx, gy = s(torch.ones(1, 1, 8, 8))
    assert gx.abs().max() < 1e-6 and gy.abs().max() < 1e-6


def test_deblur_events_bilinear_path():
    """round_idx=False (bilinear) path incl. the 4x polarity-mask repeat."""
    ev, pol = _events(B=1, N=32)
    flow = torch.rand(1, 2, 16, 16) * 0.005
    iwe = deblur_events(flow, ev, (16, 16), flow_scaling=16,
                        round_idx=False, polarity_mask=pol[:, :, 0:1])
    assert iwe.shape == (1, 1, 16, 16)
    # bilinear weights of in-range positive events sum to <= n_pos
    n_pos = pol[:, :, 0].sum().item()
    assert 0 < iwe.sum().item() <= n_pos + 1e-4


def test_averaged_iwe_divides_by_distinct_sources():
    """Two source pixels warped onto one destination: the averaged IWE is
    raw_count / n_distinct_sources (parity: ESR:loss/flow.py:113-232)."""
    H = W = 8
    scaling = max(H, W)
    # events at (y=0,x=0) and (y=0,x=2), both positive, ts=0
    ev = torch.tensor([[[0.0, 0.0, 0.0, 1.0],
                        [0.0, 0.0, 2.0, 1.0]]])   # (ts, y, x, p)
    pol = torch.tensor([[[1.0, 0.0], [1.0, 0.0]]])
    flow = torch.zeros(1, 2, H, W)
    flow[0, 0, 0, 0] = 1.0 / scaling    # x-flow at (0,0): +1 px
    flow[0, 0, 0, 2] = -1.0 / scaling   # x-flow at (0,2): -1 px
    avg = AveragedIWE((H, W))
    out = avg(flow, ev, pol)
    assert out[0, 0, 0, 1].item() == 1.0      # 2 events / 2 sources
    assert out[0, 0].sum().item() == 1.0
    # same flow for both events from ONE source pixel -> no averaging
    ev2 = torch.tensor([[[0.0, 0.0, 0.0, 1.0],
                         [0.0, 0.0, 0.0, 1.0]]])
    out2 = avg(flow, ev2, pol)
    assert out2[0, 0, 0, 1].item() == 2.0     # 2 events / 1 source
