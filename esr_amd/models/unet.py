"""Recurrent U-Net model family (parity: ESR:models/unet.py:19-498, the
e2vid-heritage alternate models: BaseUNet / UNetRecurrent / SRUNetRecurrent /
MultiResUNet).  Not used by the flagship config but part of the framework's
model zoo; built from esr_amd blocks (fused ConvGRU cells, selectable
pixel-shuffle upsamplers)."""

from __future__ import annotations

import torch
import torch.nn as nn


from .blocks import (ConvLayer, PixelShuffleUpsample, RecurrentConvLayer,
                     ResidualBlock, TransposedConvLayer, UpsampleConvLayer)
from .registry import register_model

__all__ = ["BaseUNet", "UNetRecurrent", "SRUNetRecurrent", "MultiResUNet"]


def skip_sum(x, y):
    return x + y


def skip_concat(x, y):
    return torch.cat([x, y], dim=1)


_SKIPS = {"sum": skip_sum, "concat": skip_concat}
_UPS = {"bilinear": UpsampleConvLayer, "pixelshuffle": PixelShuffleUpsample,
        "transposed": TransposedConvLayer}


class BaseUNet(nn.Module):
    """Symmetric U-Net skeleton (parity: ESR:models/unet.py:19-111)."""

    def __init__(self, base_num_channels=32, num_encoders=3,
                 num_residual_blocks=2, num_output_channels=1,
                 skip_type="sum", norm=None, use_upsample_conv=True,
                 num_bins=5, recurrent_block_type="convlstm", kernel_size=5,
                 channel_multiplier=2, upsampler=None, activation="relu"):
        super().__init__()
        self.base_num_channels = base_num_channels
        self.num_encoders = num_encoders
        self.num_residual_blocks = num_residual_blocks
        self.num_output_channels = num_output_channels
        self.kernel_size = kernel_size
        self.skip_type = skip_type
        self.skip_ftn = _SKIPS[skip_type]
        self.norm = norm
        self.num_bins = num_bins
        self.recurrent_block_type = recurrent_block_type
        self.activation = activation
        if upsampler is None:
            upsampler = "bilinear" if use_upsample_conv else "transposed"
        self.UpsampleLayer = _UPS[upsampler]

        self.encoder_input_sizes = [
            base_num_channels * channel_multiplier ** i
            for i in range(num_encoders)]
        self.encoder_output_sizes = [
            base_num_channels * channel_multiplier ** (i + 1)
            for i in range(num_encoders)]
        self.max_num_channels = self.encoder_output_sizes[-1]

    def build_encoders(self, first_in=None):
        encoders = nn.ModuleList()
        for i, (cin, cout) in enumerate(zip(self.encoder_input_sizes,
                                            self.encoder_output_sizes)):
            if i == 0 and first_in is not None:
                cin = first_in
            encoders.append(ConvLayer(cin, cout, self.kernel_size, stride=2,
                                      padding=self.kernel_size // 2,
                                      activation=self.activation,
                                      norm=self.norm))
        return encoders

    def build_recurrent_encoders(self):
        encoders = nn.ModuleList()
        for cin, cout in zip(self.encoder_input_sizes,
                             self.encoder_output_sizes):
            encoders.append(RecurrentConvLayer(
                cin, cout, self.kernel_size, stride=2,
                padding=self.kernel_size // 2,
                recurrent_block_type=self.recurrent_block_type,
                norm=self.norm))
        return encoders

    def build_resblocks(self):
        return nn.ModuleList([
            ResidualBlock(self.max_num_channels, self.max_num_channels,
                          norm=self.norm)
            for _ in range(self.num_residual_blocks)])

    def build_decoders(self, scales=None):
        ins = list(reversed(self.encoder_output_sizes))
        outs = list(reversed(self.encoder_input_sizes))
        decoders = nn.ModuleList()
        for i, (cin, cout) in enumerate(zip(ins, outs)):
            kw = {}
            if scales is not None and self.UpsampleLayer is not TransposedConvLayer:
                kw["scale"] = scales[i]
            decoders.append(self.UpsampleLayer(
                cin if self.skip_type == "sum" else 2 * cin, cout,
                self.kernel_size, padding=self.kernel_size // 2,
                norm=self.norm, **kw))
        return decoders

    def build_prediction_layer(self, num_output_channels, norm=None):
        cin = self.base_num_channels if self.skip_type == "sum" \
            else 2 * self.base_num_channels
        return ConvLayer(cin, num_output_channels, 1, activation=None,
                         norm=norm)


@register_model("UNetRecurrent")
class UNetRecurrent(BaseUNet):
    """Recurrent U-Net: each encoder followed by a ConvLSTM/ConvGRU
    (parity: ESR:models/unet.py:230-301)."""

    def __init__(self, final_activation="none", **kwargs):
        super().__init__(**kwargs)
        self.final_activation = getattr(torch, final_activation, None) \
            if final_activation != "none" else None
        self.head = ConvLayer(self.num_bins, self.base_num_channels,
                              self.kernel_size, stride=1,
                              padding=self.kernel_size // 2)
        self.encoders = self.build_recurrent_encoders()
        self.resblocks = self.build_resblocks()
        self.decoders = self.build_decoders()
        self.pred = self.build_prediction_layer(self.num_output_channels,
                                                self.norm)
        self.states = [None] * self.num_encoders

    def reset_states(self):
        self.states = [None] * self.num_encoders

    def forward(self, x):
        x = self.head(x)
        head = x
        blocks = []
        for i, enc in enumerate(self.encoders):
            x, self.states[i] = enc(x, self.states[i])
            blocks.append(x)
        for rb in self.resblocks:
            x = rb(x)
        for i, dec in enumerate(self.decoders):
            x = dec(self.skip_ftn(x, blocks[self.num_encoders - i - 1]))
        out = self.pred(self.skip_ftn(x, head))
        if self.final_activation is not None:
            out = self.final_activation(out)
        return out


@register_model("SRUNetRecurrent")
class SRUNetRecurrent(BaseUNet):
    """Recurrent SR U-Net: first decoder upsamples 4x, remaining 2x, skips
    are 2x-upsampled before fusion -> output at 2x the input resolution
    (parity: ESR:models/unet.py:393-498).

    Divergence note: the reference combines the bottleneck with a
    2x-upsampled copy of itself before the 4x decoder, which only
    type-checks when the bottleneck is 1x1 (broadcast); its forward is
    shape-inconsistent at any other input size.  Here decoder 0 fuses at
    native resolution (a residual around the resblocks) and the remaining
    skips are 2x-upsampled, which keeps every fusion size-consistent and
    still yields a 2x output."""

    def __init__(self, final_activation="none", **kwargs):
        super().__init__(**kwargs)
        self.final_activation = getattr(torch, final_activation, None) \
            if final_activation != "none" else None
        self.head = ConvLayer(self.num_bins, self.base_num_channels,
                              self.kernel_size, stride=1,
                              padding=self.kernel_size // 2)
        self.encoders = self.build_recurrent_encoders()
        self.resblocks = self.build_resblocks()
        scales = [4] + [2] * (self.num_encoders - 1)
        self.decoders = self.build_decoders(scales=scales)
        self.skip_upsampler = self._build_skip_upsampler()
        self.pred = self.build_prediction_layer(self.num_output_channels,
                                                self.norm)
        self.states = [None] * self.num_encoders

    def _build_skip_upsampler(self):
        skip_sizes = self.encoder_output_sizes[::-1] + [self.base_num_channels]
        ups = nn.ModuleList()
        for cin in skip_sizes:
            kw = {} if self.UpsampleLayer is TransposedConvLayer else {"scale": 2}
            ups.append(self.UpsampleLayer(
                cin, cin, self.kernel_size,
                padding=self.kernel_size // 2, norm=self.norm, **kw))
        return ups

    def reset_states(self):
        self.states = [None] * self.num_encoders

    def forward(self, x):
        x = self.head(x)
        head = x
        blocks = []
        for i, enc in enumerate(self.encoders):
            x, self.states[i] = enc(x, self.states[i])
            blocks.append(x)
        for rb in self.resblocks:
            x = rb(x)
        for i, dec in enumerate(self.decoders):
            blk = blocks[self.num_encoders - i - 1]
            skip = blk if i == 0 else self.skip_upsampler[i](blk)
            x = dec(self.skip_ftn(x, skip))
        out = self.pred(self.skip_ftn(x, self.skip_upsampler[-1](head)))
        if self.final_activation is not None:
            out = self.final_activation(out)
        return out


@register_model("MultiResUNet")
class MultiResUNet(BaseUNet):
    """U-Net with a prediction at every decoder scale, fed forward as a
    concat skip (parity: ESR:models/unet.py:304-390)."""

    def __init__(self, final_activation="none", **kwargs):
        kwargs["skip_type"] = "concat"
        super().__init__(**kwargs)
        self.final_activation = getattr(torch, final_activation, None) \
            if final_activation != "none" else None
        self.encoders = self.build_encoders(first_in=self.num_bins)
        self.resblocks = self.build_resblocks()
        ins = list(reversed(self.encoder_output_sizes))
        outs = list(reversed(self.encoder_input_sizes))
        self.decoders = nn.ModuleList()
        self.preds = nn.ModuleList()
        for i, (cin, cout) in enumerate(zip(ins, outs)):
            extra = self.num_output_channels if i > 0 else 0
            self.decoders.append(self.UpsampleLayer(
                2 * cin + extra, cout, self.kernel_size,
                padding=self.kernel_size // 2, norm=self.norm))
            self.preds.append(ConvLayer(cout, self.num_output_channels, 1,
                                        activation=None, norm=self.norm))

    def forward(self, x):
        blocks = []
        for enc in self.encoders:
            x = enc(x)
            blocks.append(x)
        for rb in self.resblocks:
            x = rb(x)
        predictions = []
        for i, (dec, pred) in enumerate(zip(self.decoders, self.preds)):
            skip = blocks[self.num_encoders - i - 1]
            inp = torch.cat([x, skip], dim=1)
            if i > 0:
                # decoder i-1's prediction is already at skip_i's resolution
                inp = torch.cat([inp, predictions[-1]], dim=1)
            x = dec(inp)
            p = pred(x)
            if self.final_activation is not None:
                p = self.final_activation(p)
            predictions.append(p)
        return predictions
