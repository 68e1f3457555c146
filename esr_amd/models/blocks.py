"""NN building blocks (parity: ESR:models/submodules.py).

Kept lean: only the blocks the ESR model family actually uses, plus the
pixel-shuffle upsampler the MI355X redesign prefers for the decoder
(sub-pixel conv has no interpolate round-trip through HBM; the reference
uses bilinear-interp + conv, ESR:models/submodules.py:254-299 — both are
available and config-selectable).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.conv import conv2d_act
from ..ops.convgru import ConvGRUCell

__all__ = [
    "ConvLayer",
    "ResidualBlock",
    "UpsampleConvLayer",
    "PixelShuffleUpsample",
    "TransposedConvLayer",
    "RecurrentConvLayer",
    "ConvLSTMCell",
    "MLP",
]

_ACTS = {
    "relu": F.relu,
    "sigmoid": torch.sigmoid,
    "tanh": torch.tanh,
    None: None,
}


def _norm_layer(norm, ch, bn_momentum=0.1):
    if norm == "BN":
        return nn.BatchNorm2d(ch, momentum=bn_momentum)
    if norm == "IN":
        return nn.InstanceNorm2d(ch, track_running_stats=True)
    return None


class ConvLayer(nn.Module):
    """Conv2d + optional norm + optional activation
    (parity: ESR:models/submodules.py:159-200)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, activation="relu", norm=None, bn_momentum=0.1):
        super().__init__()
        bias = norm != "BN"
        self.conv2d = nn.Conv2d(in_channels, out_channels, kernel_size,
                                stride, padding, bias=bias)
        self.activation = _ACTS[activation]
        self._act_name = activation
        self.norm_layer = _norm_layer(norm, out_channels, bn_momentum)

    def forward(self, x):
        if self.norm_layer is None:
            # native gfx950 conv with bias+act fused into the epilogue
            out = conv2d_act(x, self.conv2d, self._act_name)
            if out is not None:
                return out
        out = self.conv2d(x)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class ResidualBlock(nn.Module):
    """3x3-3x3 residual block (parity: ESR:models/submodules.py:347-409)."""

    def __init__(self, in_channels, out_channels, stride=1, downsample=None,
                 norm=None, bn_momentum=0.1, final_activation=True):
        super().__init__()
        bias = norm != "BN"
        self.conv1 = nn.Conv2d(in_channels, out_channels, 3, stride, 1, bias=bias)
        self.conv2 = nn.Conv2d(out_channels, out_channels, 3, 1, 1, bias=bias)
        self.n1 = _norm_layer(norm, out_channels, bn_momentum)
        self.n2 = _norm_layer(norm, out_channels, bn_momentum)
        self.downsample = downsample
        self.final_activation = final_activation

    def forward(self, x):
        residual = x if self.downsample is None else self.downsample(x)
        if self.n1 is None:
            out = conv2d_act(x, self.conv1, "relu")
            if out is None:
                out = F.relu(self.conv1(x))
        else:
            out = F.relu(self.n1(self.conv1(x)))
        if self.n2 is None:
            out2 = conv2d_act(out, self.conv2, None)
            if out2 is None:
                out2 = self.conv2(out)
        else:
            out2 = self.n2(self.conv2(out))
        out = out2 + residual
        if self.final_activation:
            out = F.relu(out)
        return out


class UpsampleConvLayer(nn.Module):
    """Bilinear 2x upsample + conv (parity: ESR:models/submodules.py:254-299)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, activation="relu", norm=None, scale=2):
        super().__init__()
        bias = norm != "BN"
        self.conv2d = nn.Conv2d(in_channels, out_channels, kernel_size,
                                stride, padding, bias=bias)
        self.activation = _ACTS[activation]
        self._act_name = activation
        self.norm_layer = _norm_layer(norm, out_channels)
        self.scale = scale

    def forward(self, x):
        x = F.interpolate(x, scale_factor=self.scale, mode="bilinear",
                          align_corners=False)
        if self.norm_layer is None:
            out = conv2d_act(x, self.conv2d, self._act_name)
            if out is not None:
                return out
        out = self.conv2d(x)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class PixelShuffleUpsample(nn.Module):
    """Sub-pixel (pixel-shuffle) 2x upsampler: conv to 4*out then shuffle.

    MI355X-preferred decoder upsampler — the conv runs at LOW resolution
    (1/4 the pixels of post-interp conv) and the shuffle is a pure layout
    op, so HBM traffic is ~2.5x lower than bilinear-interp + conv at equal
    receptive field.  ICNR-style init keeps it equivalent to a bilinear
    start.
    """

    def __init__(self, in_channels, out_channels, kernel_size=3, stride=1,
                 padding=1, activation="relu", norm=None, scale=2):
        super().__init__()
        bias = norm != "BN"
        self.scale = scale
        self.conv2d = nn.Conv2d(in_channels, out_channels * scale * scale,
                                kernel_size, stride, padding, bias=bias)
        self.activation = _ACTS[activation]
        self.norm_layer = _norm_layer(norm, out_channels)
        self._icnr_init(out_channels)

    def _icnr_init(self, out_channels):
        # replicate a [out, in, k, k] kernel across the r^2 shuffle slots so
        # the initial output is spatially smooth (checkerboard-free).
        w = self.conv2d.weight.data
        r2 = self.scale * self.scale
        sub = torch.empty(out_channels, w.size(1), w.size(2), w.size(3))
        nn.init.kaiming_uniform_(sub, a=5 ** 0.5)
        self.conv2d.weight.data.copy_(
            sub.repeat_interleave(r2, dim=0).reshape_as(w))

    def forward(self, x):
        pre = conv2d_act(x, self.conv2d, None)
        if pre is None:
            pre = self.conv2d(x)
        out = F.pixel_shuffle(pre, self.scale)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class TransposedConvLayer(nn.Module):
    """Stride-2 deconv (parity: ESR:models/submodules.py:203-251)."""

    def __init__(self, in_channels, out_channels, kernel_size, padding=0,
                 activation="relu", norm=None):
        super().__init__()
        bias = norm != "BN"
        self.transposed_conv2d = nn.ConvTranspose2d(
            in_channels, out_channels, kernel_size, stride=2, padding=padding,
            output_padding=1, bias=bias)
        self.activation = _ACTS[activation]
        self.norm_layer = _norm_layer(norm, out_channels)

    def forward(self, x):
        out = self.transposed_conv2d(x)
        if self.norm_layer is not None:
            out = self.norm_layer(out)
        if self.activation is not None:
            out = self.activation(out)
        return out


class ConvLSTMCell(nn.Module):
    """Convolutional LSTM cell (parity: ESR:models/submodules.py:412-471).
    Single 4C-output gate conv; state is (hidden, cell)."""

    def __init__(self, input_size, hidden_size, kernel_size):
        super().__init__()
        pad = kernel_size // 2
        self.hidden_size = hidden_size
        self.gates = nn.Conv2d(input_size + hidden_size, 4 * hidden_size,
                               kernel_size, padding=pad)

    def forward(self, x, state):
        if state is None:
            z = torch.zeros(x.size(0), self.hidden_size, x.size(2), x.size(3),
                            dtype=x.dtype, device=x.device)
            state = (z, z)
        h_prev, c_prev = state
        xh = torch.cat([x, h_prev], dim=1)
        g = conv2d_act(xh, self.gates, None)
        if g is None:
            g = self.gates(xh)
        i, f, o, c_hat = torch.chunk(g, 4, dim=1)
        i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
        c = f * c_prev + i * torch.tanh(c_hat)
        h = o * torch.tanh(c)
        return h, c


class RecurrentConvLayer(nn.Module):
    """Conv followed by a recurrent block (parity:
    ESR:models/submodules.py:302-344).  Returns (output, new_state)."""

    def __init__(self, in_channels, out_channels, kernel_size=3, stride=1,
                 padding=0, recurrent_block_type="convgru", activation="relu",
                 norm=None, bn_momentum=0.1):
        super().__init__()
        assert recurrent_block_type in ("convlstm", "convgru")
        self.recurrent_block_type = recurrent_block_type
        self.conv = ConvLayer(in_channels, out_channels, kernel_size, stride,
                              padding, activation, norm, bn_momentum)
        if recurrent_block_type == "convgru":
            self.recurrent_block = ConvGRUCell(out_channels, out_channels, 3)
        else:
            self.recurrent_block = ConvLSTMCell(out_channels, out_channels, 3)

    def forward(self, x, prev_state):
        x = self.conv(x)
        if self.recurrent_block_type == "convgru":
            state = self.recurrent_block(x, prev_state)
            return state, state
        h, c = self.recurrent_block(x, prev_state)
        return h, (h, c)


class MLP(nn.Module):
    """Per-position MLP (parity: ESR:models/submodules.py:67-78)."""

    def __init__(self, input_dim, hidden_dim, output_dim, num_layers):
        super().__init__()
        self.num_layers = num_layers
        h = [hidden_dim] * (num_layers - 1)
        self.layers = nn.ModuleList(
            nn.Linear(n, k) for n, k in zip([input_dim] + h, h + [output_dim]))

    def forward(self, x):
        for i, layer in enumerate(self.layers):
            x = F.relu(layer(x)) if i < self.num_layers - 1 else layer(x)
        return x


def recursive_clone(tensor_or_state):
    """Deep-clone nested tensors/tuples/lists (parity:
    ESR:models/model_util.py:208-229 copy_states/recursive_clone)."""
    if isinstance(tensor_or_state, torch.Tensor):
        return tensor_or_state.clone()
    if isinstance(tensor_or_state, (tuple, list)):
        return type(tensor_or_state)(recursive_clone(t)
                                     for t in tensor_or_state)
    return tensor_or_state


def copy_states(states):
    """Snapshot recurrent states for later restoration."""
    return recursive_clone(states)
