"""Frame-rate upsampling by optical-flow interpolation (Super-SloMo class).

The reference vendors Super-SloMo to raise the frame rate of source videos
before event simulation (ESR:generate_dataset/upsampling/utils/model.py:
139-251, upsampler.py:100-210).  This is a compact re-implementation with
the framework's blocks: a flow UNet (F_0_1, F_1_0), an arbitrary-time flow
refinement UNet with visibility maps, backwarp via grid_sample, and a
recursive 2x upsampler utility.  Weights are random-init (no network);
the module is trainable with the photometric losses in esr_amd.loss.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .registry import register_model

__all__ = ["InterpUNet", "FrameInterpolator", "backwarp", "upsample_frames"]


def backwarp(img: torch.Tensor, flow: torch.Tensor) -> torch.Tensor:
    """Warp img by flow (pixels): out(x) = img(x + flow(x))
    (parity: ESR:generate_dataset/upsampling/utils/model.py:215-251)."""
    B, _, H, W = img.shape
    gy, gx = torch.meshgrid(
        torch.arange(H, device=img.device, dtype=img.dtype),
        torch.arange(W, device=img.device, dtype=img.dtype), indexing="ij")
    x = gx.unsqueeze(0) + flow[:, 0]
    y = gy.unsqueeze(0) + flow[:, 1]
    x = 2 * x / (W - 1) - 1
    y = 2 * y / (H - 1) - 1
    grid = torch.stack([x, y], dim=3)
    return F.grid_sample(img, grid, mode="bilinear", padding_mode="border",
                         align_corners=True)


class _Down(nn.Module):
    def __init__(self, cin, cout, k):
        super().__init__()
        self.c1 = nn.Conv2d(cin, cout, k, padding=k // 2)
        self.c2 = nn.Conv2d(cout, cout, k, padding=k // 2)

    def forward(self, x):
        x = F.avg_pool2d(x, 2)
        x = F.leaky_relu(self.c1(x), 0.1)
        return F.leaky_relu(self.c2(x), 0.1)


class _Up(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.c1 = nn.Conv2d(cin, cout, 3, padding=1)
        self.c2 = nn.Conv2d(2 * cout, cout, 3, padding=1)

    def forward(self, x, skip):
        x = F.interpolate(x, scale_factor=2, mode="bilinear",
                          align_corners=False)
        x = F.leaky_relu(self.c1(x), 0.1)
        x = F.leaky_relu(self.c2(torch.cat([x, skip], dim=1)), 0.1)
        return x


@register_model("InterpUNet")
class InterpUNet(nn.Module):
    """6-level leaky-ReLU UNet used for both interpolation stages."""

    def __init__(self, in_channels, out_channels, base=32):
        super().__init__()
        self.head1 = nn.Conv2d(in_channels, base, 7, padding=3)
        self.head2 = nn.Conv2d(base, base, 7, padding=3)
        chs = [base, 2 * base, 4 * base, 8 * base, 16 * base, 16 * base]
        ks = [5, 3, 3, 3, 3]
        self.downs = nn.ModuleList(
            _Down(chs[i], chs[i + 1], ks[i]) for i in range(5))
        self.ups = nn.ModuleList(
            _Up(chs[5 - i], chs[4 - i]) for i in range(5))
        self.tail = nn.Conv2d(base, out_channels, 3, padding=1)

    def forward(self, x):
        x = F.leaky_relu(self.head1(x), 0.1)
        x = F.leaky_relu(self.head2(x), 0.1)
        skips = [x]
        for d in self.downs:
            x = d(x)
            skips.append(x)
        for i, u in enumerate(self.ups):
            x = u(x, skips[4 - i])
        return self.tail(x)


@register_model("FrameInterpolator")
class FrameInterpolator(nn.Module):
    """Two-stage arbitrary-time frame interpolation."""

    def __init__(self, base=32):
        super().__init__()
        self.flow_net = InterpUNet(2, 4, base)          # grayscale pair -> flows
        self.refine_net = InterpUNet(8, 5, base)  # I0,I1,g0,g1,Ft0,Ft1

    def forward(self, I0, I1, t: float, return_flows: bool = False):
        flows = self.flow_net(torch.cat([I0, I1], dim=1))
        F01 = flows[:, 0:2]
        F10 = flows[:, 2:4]
        Ft0_hat = -(1 - t) * t * F01 + t * t * F10
        Ft1_hat = (1 - t) * (1 - t) * F01 - t * (1 - t) * F10
        g0 = backwarp(I0, Ft0_hat)
        g1 = backwarp(I1, Ft1_hat)
        ref = self.refine_net(torch.cat(
            [I0, I1, g0, g1, Ft0_hat, Ft1_hat], dim=1))
        Ft0 = Ft0_hat + ref[:, 0:2]
        Ft1 = Ft1_hat + ref[:, 2:4]
        V0 = torch.sigmoid(ref[:, 4:5])
        V1 = 1 - V0
        g0 = backwarp(I0, Ft0)
        g1 = backwarp(I1, Ft1)
        num = (1 - t) * V0 * g0 + t * V1 * g1
        den = (1 - t) * V0 + t * V1
        out = num / (den + 1e-8)
        if return_flows:  # for self-training (tools/train_interp.py)
            return out, (F01, F10, Ft0, Ft1)
        return out


@torch.no_grad()
def upsample_frames(model: FrameInterpolator, frames: torch.Tensor,
                    factor: int = 2) -> torch.Tensor:
    """Recursively double the frame rate `log2(factor)` times
    (parity: ESR:generate_dataset/upsampling/utils/upsampler.py:160-210)."""
    assert factor & (factor - 1) == 0, "factor must be a power of 2"
    while factor > 1:
        out = []
        for i in range(frames.shape[0] - 1):
            I0 = frames[i:i + 1]
            I1 = frames[i + 1:i + 2]
            out.append(I0)
            out.append(model(I0, I1, 0.5))
        out.append(frames[-1:])
        frames = torch.cat(out, dim=0)
        factor //= 2
    return frames
