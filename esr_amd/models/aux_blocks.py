"""Auxiliary blocks from the reference's submodule zoo
(ESR:models/submodules.py:9-155, :518-752): attention, 1D/3D conv layers,
inception/dilated aggregation, and point-cloud feature ops (KNN graph +
DenseEdgeConv).  Kept for model-zoo parity; the flagship ESRNet does not
use them."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["SelfAttention", "ConvLayer1D", "ConvLayer3D", "InceptionBlock",
           "DilatedBlock", "knn", "DenseEdgeConv", "FeedbackBlock"]


class SelfAttention(nn.Module):
    """Offset self-attention over point/sequence features
    (parity: ESR:models/submodules.py:80-112)."""

    def __init__(self, channels):
        super().__init__()
        self.q_conv = nn.Conv1d(channels, channels // 4, 1, bias=False)
        self.k_conv = nn.Conv1d(channels, channels // 4, 1, bias=False)
        self.q_conv.weight = self.k_conv.weight  # shared QK projection
        self.v_conv = nn.Conv1d(channels, channels, 1)
        self.trans_conv = nn.Conv1d(channels, channels, 1)
        self.after_norm = nn.BatchNorm1d(channels)

    def forward(self, x):
        """x: [B, N, C] -> [B, N, C]"""
        x = x.transpose(1, 2)
        q = self.q_conv(x).permute(0, 2, 1)
        k = self.k_conv(x)
        v = self.v_conv(x)
        attention = torch.softmax(q @ k, dim=-1)
        attention = attention / (1e-9 + attention.sum(dim=1, keepdim=True))
        x_r = v @ attention
        x_r = F.relu(self.after_norm(self.trans_conv(x - x_r)))
        return (x + x_r).transpose(1, 2)


class ConvLayer1D(nn.Module):
    """1D conv + optional BN + activation (parity:
    ESR:models/submodules.py:115-155)."""

    def __init__(self, in_channels, out_channels, kernel_size=1, stride=1,
                 padding=0, activation="relu", norm=None):
        super().__init__()
        bias = norm != "BN"
        self.conv = nn.Conv1d(in_channels, out_channels, kernel_size, stride,
                              padding, bias=bias)
        self.norm_layer = nn.BatchNorm1d(out_channels) if norm == "BN" else None
        self.activation = getattr(torch, activation) if activation else None

    def forward(self, x):
        out = self.conv(x)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class ConvLayer3D(nn.Module):
    """3D conv + optional BN + activation (parity:
    ESR:models/submodules.py:518-565)."""

    def __init__(self, in_channels, out_channels, kernel_size=3, stride=1,
                 padding=1, activation="relu", norm=None):
        super().__init__()
        bias = norm != "BN"
        self.conv = nn.Conv3d(in_channels, out_channels, kernel_size, stride,
                              padding, bias=bias)
        self.norm_layer = nn.BatchNorm3d(out_channels) if norm == "BN" else None
        self.activation = getattr(torch, activation) if activation else None

    def forward(self, x):
        out = self.conv(x)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class InceptionBlock(nn.Module):
    """Parallel 1/3/5 kernel branches summed (parity:
    ESR:models/submodules.py:9-35)."""

    def __init__(self, in_channels, out_channels, norm=None):
        super().__init__()
        self.branches = nn.ModuleList([
            nn.Conv2d(in_channels, out_channels, k, padding=k // 2)
            for k in (1, 3, 5)])

    def forward(self, x):
        out = 0
        for b in self.branches:
            out = out + b(x)
        return F.relu(out)


class DilatedBlock(nn.Module):
    """Parallel dilation 1/2/4 3x3 branches summed (parity:
    ESR:models/submodules.py:38-63)."""

    def __init__(self, in_channels, out_channels, norm=None):
        super().__init__()
        self.branches = nn.ModuleList([
            nn.Conv2d(in_channels, out_channels, 3, padding=d, dilation=d)
            for d in (1, 2, 4)])

    def forward(self, x):
        out = 0
        for b in self.branches:
            out = out + b(x)
        return F.relu(out)


def knn(pos: torch.Tensor, k: int) -> torch.Tensor:
    """k-nearest-neighbour indices over point positions [B, N, D]
    (parity: ESR:models/submodules.py:626-660).  Returns [B, N, k]."""
    dist = torch.cdist(pos, pos)
    return dist.topk(k, dim=-1, largest=False).indices


class DenseEdgeConv(nn.Module):
    """Densely-connected edge convolution over a KNN graph
    (parity: ESR:models/submodules.py:663-752)."""

    def __init__(self, in_channels, growth, num_layers=3, k=16):
        super().__init__()
        self.k = k
        self.layers = nn.ModuleList()
        ch = 2 * in_channels
        for _ in range(num_layers):
            self.layers.append(nn.Conv2d(ch, growth, 1))
            ch = ch + growth

    def forward(self, feats: torch.Tensor, pos: torch.Tensor | None = None):
        """feats: [B, N, C]; pos defaults to feats. Returns [B, N, C_out]."""
        B, N, C = feats.shape
        idx = knn(pos if pos is not None else feats, self.k)     # [B,N,k]
        gather = feats.unsqueeze(1).expand(B, N, N, C).gather(
            2, idx.unsqueeze(-1).expand(B, N, self.k, C))        # [B,N,k,C]
        center = feats.unsqueeze(2).expand_as(gather)
        edge = torch.cat([center, gather - center], dim=-1)      # [B,N,k,2C]
        x = edge.permute(0, 3, 1, 2)                             # [B,2C,N,k]
        for conv in self.layers:
            y = F.relu(conv(x))
            x = torch.cat([x, y], dim=1)
        return x.max(dim=-1).values.transpose(1, 2)              # [B,N,C_out]


class FeedbackBlock(nn.Module):
    """SRFBN-style feedback block: iterative up/down projection pairs with
    dense connections (parity: ESR:models/submodules.py:755-871, the SRFBN
    helper family used as an SR baseline)."""

    def __init__(self, channels, num_groups=3, scale=2):
        super().__init__()
        k, s, p = {2: (6, 2, 2), 4: (8, 4, 2), 8: (12, 8, 2)}[scale]
        self.compress_in = nn.Conv2d(2 * channels, channels, 1)
        self.up_blocks = nn.ModuleList()
        self.down_blocks = nn.ModuleList()
        self.uptran = nn.ModuleList()
        self.downtran = nn.ModuleList()
        for i in range(num_groups):
            self.up_blocks.append(
                nn.ConvTranspose2d(channels, channels, k, s, p))
            self.down_blocks.append(nn.Conv2d(channels, channels, k, s, p))
            if i > 0:
                self.uptran.append(nn.Conv2d((i + 1) * channels, channels, 1))
                self.downtran.append(nn.Conv2d((i + 1) * channels, channels, 1))
        self.compress_out = nn.Conv2d(num_groups * channels, channels, 1)
        self.last_hidden = None

    def reset_state(self):
        self.last_hidden = None

    def forward(self, x):
        if self.last_hidden is None:
            self.last_hidden = torch.zeros_like(x)
        x = self.compress_in(torch.cat([x, self.last_hidden], dim=1))
        lows, highs = [x], []
        for i, (up, down) in enumerate(zip(self.up_blocks, self.down_blocks)):
            lo = torch.cat(lows, dim=1)
            if i > 0:
                lo = self.uptran[i - 1](lo)
            hi = F.relu(up(lo))
            highs.append(hi)
            hi_cat = torch.cat(highs, dim=1)
            if i > 0:
                hi_cat = self.downtran[i - 1](hi_cat)
            lows.append(F.relu(down(hi_cat)))
        out = self.compress_out(torch.cat(lows[1:], dim=1))
        self.last_hidden = out
        return out
