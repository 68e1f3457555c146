"""Spatial-gradient ops (parity: ESR:myutils/gradients.py:7-33)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class Sobel(nn.Module):
    """Normalized Sobel gradients with replication padding."""

    def __init__(self, device=None):
        super().__init__()
        kx = torch.tensor([[-1., 0., 1.], [-2., 0., 2.], [-1., 0., 1.]])
        ky = torch.tensor([[-1., -2., -1.], [0., 0., 0.], [1., 2., 1.]])
        self.register_buffer("kx", kx.view(1, 1, 3, 3) / 8)
        self.register_buffer("ky", ky.view(1, 1, 3, 3) / 8)
        if device is not None:
            self.to(device)

    def forward(self, x):
        x = x.reshape(-1, 1, x.shape[-2], x.shape[-1])
        x = F.pad(x, (1, 1, 1, 1), mode="replicate")
        return F.conv2d(x, self.kx), F.conv2d(x, self.ky)
