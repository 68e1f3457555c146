"""Self-supervised optical-flow losses over event streams.

Parity: ESR:loss/flow.py:15-232 — contrast-maximization (per-pixel,
per-polarity average-timestamp minimization, forward+backward) with
Charbonnier flow smoothing, and the averaged image of warped events.
Unused by supervised ESR training (the reference builds but never calls
them); part of the framework's loss zoo for flow-based training.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..utils.iwe import (event_flow_lookup, get_interpolation, interpolate)

__all__ = ["EventWarping", "AveragedIWE"]


class EventWarping(nn.Module):
    """Contrast-maximization loss (Zhu et al., CVPR'19 style)."""

    def __init__(self, flow_regul_weight: float = 1.0):
        super().__init__()
        self.weight = flow_regul_weight

    def forward(self, flow_list, event_list, pol_mask, resolution):
        """flow_list: list of [B,2,H,W]; event_list: [B,N,4] (ts,y,x,p);
        pol_mask: [B,N,2]."""
        res = resolution
        flow_scaling = max(res)
        pm4 = pol_mask.repeat(1, 4, 1)
        ts4 = event_list[:, :, 0:1].repeat(1, 4, 1)

        loss = 0
        for flow in flow_list:
            ev_flow = event_flow_lookup(flow, event_list, res)

            for tref, ts_w in ((1, ts4), (0, 1 - ts4)):
                idx, w = get_interpolation(event_list, ev_flow, tref, res,
                                           flow_scaling)
                iwe_pos = interpolate(idx.long(), w, res, pm4[:, :, 0:1])
                iwe_neg = interpolate(idx.long(), w, res, pm4[:, :, 1:2])
                ts_pos = interpolate(idx.long(), w * ts_w, res, pm4[:, :, 0:1])
                ts_neg = interpolate(idx.long(), w * ts_w, res, pm4[:, :, 1:2])
                ts_pos = ts_pos / (iwe_pos + 1e-9)
                ts_neg = ts_neg / (iwe_neg + 1e-9)
                loss = loss + ts_pos.square().sum() + ts_neg.square().sum()

            f = flow.view(flow.shape[0], 2, res[0], res[1])
            dx = torch.sqrt((f[:, :, :-1] - f[:, :, 1:]) ** 2 + 1e-6)
            dy = torch.sqrt((f[:, :, :, :-1] - f[:, :, :, 1:]) ** 2 + 1e-6)
            loss = loss + self.weight * (dx.sum() + dy.sum())
        return loss


class AveragedIWE(nn.Module):
    """Per-pixel, per-polarity AVERAGE count of warped events: each
    receiving pixel's count is divided by the number of distinct source
    pixels that contributed to it (parity: ESR:loss/flow.py:113-232)."""

    def __init__(self, resolution, device=None):
        super().__init__()
        self.res = list(resolution)
        self.flow_scaling = max(resolution)

    def forward(self, flow, event_list, pol_mask):
        res = self.res
        npix = res[0] * res[1]
        B = flow.shape[0]
        device = flow.device

        src_idx = (event_list[:, :, 1] * res[1] + event_list[:, :, 2]) \
            .long().unsqueeze(-1)
        ev_flow = event_flow_lookup(flow, event_list, res)
        fw_idx, fw_w = get_interpolation(event_list, ev_flow, 1, res,
                                         self.flow_scaling, round_idx=True)
        iwe_pos = interpolate(fw_idx.long(), fw_w, res, pol_mask[:, :, 0:1])
        iwe_neg = interpolate(fw_idx.long(), fw_w, res, pol_mask[:, :, 1:2])
        if fw_idx.shape[1] == 0:
            return torch.cat([iwe_pos, iwe_neg], dim=1)

        # polarity id: 1 = positive, 0 = negative, 2 = unfeasible mapping
        pol = (event_list[:, :, 3:4] >= 1).long()
        pol = torch.where(fw_w == 0, torch.full_like(pol, 2), pol)

        contrib = torch.zeros(B, 2, npix, device=device)
        for b in range(B):
            # unique (pol, src, dst) triples -> distinct source pixels per
            # (pol, dst)
            key = (pol[b, :, 0] * npix + src_idx[b, :, 0]) * npix \
                + fw_idx[b, :, 0].long()
            uniq = torch.unique(key)
            u_pol = uniq // (npix * npix)
            u_dst = uniq % npix
            for p_id, ch in ((1, 0), (0, 1)):
                sel = u_dst[u_pol == p_id]
                contrib[b, ch].scatter_add_(0, sel,
                                            torch.ones_like(sel, dtype=torch.float))
        contrib = contrib.view(B, 2, res[0], res[1])
        iwe = torch.cat([iwe_pos, iwe_neg], dim=1)
        return torch.where(contrib > 0, iwe / contrib, iwe)
