"""Self-supervised image-reconstruction losses (photometric constancy).

Parity: ESR:loss/reconstruction.py:17-145 — (1) event-camera generative
model error between the flow-warped image-gradient brightness increment and
the event-integrated increment, (2) temporal consistency of consecutive
reconstructions under the flow, (3) total-variation regularization.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ..utils.gradients import Sobel
from .flow import AveragedIWE

__all__ = ["BrightnessConstancy"]


class BrightnessConstancy(torch.nn.Module):
    def __init__(self, resolution, regul_weights=(0.1, 0.1), device=None):
        super().__init__()
        self.res = list(resolution)
        self.flow_scaling = max(resolution)
        self.weights = regul_weights
        self.sobel = Sobel(device)
        my, mx = torch.meshgrid(
            torch.arange(self.res[0], dtype=torch.float32),
            torch.arange(self.res[1], dtype=torch.float32), indexing="ij")
        self.register_buffer("indices", torch.stack([my, mx]).unsqueeze(0))
        self.averaged_iwe = AveragedIWE(resolution, device)
        if device is not None:
            self.to(device)

    def _warp_grid(self, flow):
        wy = self.indices[:, 0:1] - flow[:, 1:2] * self.flow_scaling
        wx = self.indices[:, 1:2] - flow[:, 0:1] * self.flow_scaling
        wy = 2 * wy / (self.res[0] - 1) - 1
        wx = 2 * wx / (self.res[1] - 1) - 1
        return torch.cat([wx, wy], dim=1).permute(0, 2, 3, 1)

    def generative_model(self, flow, img, event_cnt, event_list, pol_mask):
        """Brightness-increment (generative-model) error."""
        flow_mask = (event_cnt.sum(dim=1, keepdim=True) > 0).float()
        flow = flow * flow_mask
        grid = self._warp_grid(flow)
        gradx, grady = self.sobel(img)
        wgx = F.grid_sample(gradx, grid, mode="bilinear", padding_mode="zeros",
                            align_corners=True)
        wgy = F.grid_sample(grady, grid, mode="bilinear", padding_mode="zeros",
                            align_corners=True)
        pred_dL = (wgx * flow[:, 0:1] + wgy * flow[:, 1:2]) * self.flow_scaling
        avg_iwe = self.averaged_iwe(flow, event_list, pol_mask)
        event_dL = avg_iwe[:, 0:1] - avg_iwe[:, 1:2]
        err = (event_dL + pred_dL).flatten(2)
        return err.norm(p=2, dim=2).square().sum()

    def temporal_consistency(self, flow, prev_img, img):
        grid = self._warp_grid(flow)
        warped_prev = F.grid_sample(prev_img, grid, mode="bilinear",
                                    padding_mode="zeros", align_corners=True)
        err = (img - warped_prev).flatten(2).norm(p=1, dim=2).sum()
        return self.weights[1] * err

    def regularization(self, img):
        dx = (img[:, :, :-1] - img[:, :, 1:]).abs().sum()
        dy = (img[:, :, :, :-1] - img[:, :, :, 1:]).abs().sum()
        return self.weights[0] * (dx + dy)
