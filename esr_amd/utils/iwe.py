"""Image-of-warped-events (IWE) utilities.

Parity: ESR:myutils/iwe.py:4-151.  Events here are [B, N, 4] rows
(ts, y, x, p) like the reference's loss path; warping moves each event along
a per-event flow vector toward a reference time, and the warped events are
splatted with bilinear (or rounded) weights.
"""

from __future__ import annotations

import torch

__all__ = ["purge_unfeasible", "get_interpolation", "interpolate",
           "deblur_events", "compute_pol_iwe", "event_flow_lookup"]


def purge_unfeasible(x: torch.Tensor, res):
    """Zero out-of-image warped locations; returns (masked locations, mask)."""
    mask = ((x[:, :, 0:1] >= 0) & (x[:, :, 0:1] < res[0]) &
            (x[:, :, 1:2] >= 0) & (x[:, :, 1:2] < res[1])).float()
    return x * mask, mask


def event_flow_lookup(flow: torch.Tensor, event_list: torch.Tensor, res):
    """Gather the (y, x) flow vector at each event's integer location.

    flow: [B, 2, H, W] (channel 0 = x, 1 = y like the reference);
    event_list rows are (ts, y, x, p).  Returns [B, N, 2] (flow_y, flow_x).
    """
    idx = (event_list[:, :, 1] * res[1] + event_list[:, :, 2]).long()
    flat = flow.reshape(flow.shape[0], 2, -1)
    fy = torch.gather(flat[:, 1], 1, idx)
    fx = torch.gather(flat[:, 0], 1, idx)
    return torch.stack([fy, fx], dim=2)


def get_interpolation(events, flow, tref, res, flow_scaling, round_idx=False):
    """Warp events to tref along per-event flow; return flat scatter indices
    and interpolation weights (parity: ESR:myutils/iwe.py:20-75)."""
    warped = events[:, :, 1:3] + (tref - events[:, :, 0:1]) * flow * flow_scaling
    if round_idx:
        idx = torch.round(warped)
        weights = torch.ones_like(idx)
    else:
        top_y = torch.floor(warped[:, :, 0:1])
        left_x = torch.floor(warped[:, :, 1:2])
        corners = [torch.cat([top_y + dy, left_x + dx], dim=2)
                   for dy in (0, 1) for dx in (0, 1)]
        idx = torch.cat(corners, dim=1)
        rep = warped.repeat(1, 4, 1)
        weights = torch.clamp(1 - torch.abs(rep - idx), min=0)
    idx, mask = purge_unfeasible(idx, res)
    weights = torch.prod(weights, dim=-1, keepdim=True) * mask
    flat = (idx[:, :, 0] * res[1] + idx[:, :, 1]).unsqueeze(-1)
    return flat, weights


def interpolate(idx, weights, res, polarity_mask=None):
    """Scatter warped-event weights into an image [B, 1, H, W]
    (parity: ESR:myutils/iwe.py:78-94)."""
    if polarity_mask is not None:
        weights = weights * polarity_mask
    iwe = torch.zeros(idx.shape[0], res[0] * res[1], 1, device=idx.device)
    iwe.scatter_add_(1, idx.long(), weights)
    return iwe.view(idx.shape[0], 1, res[0], res[1])


def deblur_events(flow, event_list, res, flow_scaling=128, round_idx=True,
                  polarity_mask=None):
    """Motion-compensate events with a dense flow map and splat
    (parity: ESR:myutils/iwe.py:97-128)."""
    ev_flow = event_flow_lookup(flow, event_list, res)
    fw_idx, fw_weights = get_interpolation(event_list, ev_flow, 1, res,
                                           flow_scaling, round_idx=round_idx)
    if not round_idx and polarity_mask is not None:
        polarity_mask = polarity_mask.repeat(1, 4, 1)
    return interpolate(fw_idx.long(), fw_weights, res,
                       polarity_mask=polarity_mask)


def compute_pol_iwe(flow, event_list, res, pos_mask, neg_mask,
                    flow_scaling=128, round_idx=True):
    """Per-polarity IWE [B, 2, H, W] (parity: ESR:myutils/iwe.py:131-151)."""
    pos = deblur_events(flow, event_list, res, flow_scaling, round_idx, pos_mask)
    neg = deblur_events(flow, event_list, res, flow_scaling, round_idx, neg_mask)
    return torch.cat([pos, neg], dim=1)
