"""ESRNet — the recurrent event-stream super-resolution network.

Architecture parity with the reference's ``DeepRecurrNet``
(ESR:models/model.py:294-344): head conv -> 3-stage stride-2 encoder ->
temporal propagation (local triplet gating + bi-directional ConvGRU with
persistent state across forward calls = the BPTT link) -> spatio-temporal
fusion (deformable alignment to the centre frame + attention + 3-level
decoder) -> tail conv, with /8 padding and output crop.

MI355X-first deltas (math-preserving unless noted):
  * the bi-directional GRU scan batches the forward and reverse direction
    into one conv call per step (2x fewer kernel launches, same math);
  * the decoder upsampler is config-selectable: 'bilinear' (reference
    parity, interp+conv) or 'pixelshuffle' (sub-pixel conv; ~2.5x less HBM
    traffic — the default for benchmarks; changes the parametrization, not
    the interface);
  * deformable alignment runs on the hand-written gfx950 HIP kernels
    (esr_amd.ops.dcn).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.dcn import DeformAlign2d
from .blocks import (ConvLayer, MLP, RecurrentConvLayer, ResidualBlock,
                     UpsampleConvLayer, PixelShuffleUpsample)
from .registry import register_model

__all__ = ["ESRNet"]


def _pad_multiple(x: torch.Tensor, mult: int):
    """Centre-pad the trailing H,W dims to a multiple of `mult`.

    Parity: ESR:models/model_util.py:133-164 (CropSize pad/crop: ceil on
    top/left).  Returns (padded, (top, left, H, W)).
    """
    H, W = x.shape[-2:]
    Hc = math.ceil(H / mult) * mult
    Wc = math.ceil(W / mult) * mult
    if Hc == H and Wc == W:
        return x, None
    top = math.ceil((Hc - H) / 2)
    bottom = Hc - H - top
    left = math.ceil((Wc - W) / 2)
    right = Wc - W - left
    return F.pad(x, (left, right, top, bottom)), (top, left, H, W)


def _crop(x: torch.Tensor, box):
    if box is None:
        return x
    top, left, H, W = box
    return x[..., top:top + H, left:left + W].contiguous()


class FeatsExtract(nn.Module):
    """3x stride-2 conv pyramid; returns features deepest-first
    (parity: ESR:models/model.py:20-45)."""

    def __init__(self, basech=16, norm=None, activation="relu"):
        super().__init__()
        self.convblock = nn.ModuleList([
            ConvLayer(basech, 2 * basech, 3, 2, 1, activation, norm),
            ConvLayer(2 * basech, 4 * basech, 3, 2, 1, activation, norm),
            ConvLayer(4 * basech, 8 * basech, 3, 2, 1, activation, norm),
        ])

    def forward(self, x):
        outs = []
        for blk in self.convblock:
            x = blk(x)
            outs.append(x)
        return outs[::-1]


class TimePropagation(nn.Module):
    """Local (triplet gating) + global (bi-dir ConvGRU) temporal correlation
    with persistent hidden state (parity: ESR:models/model.py:48-153)."""

    def __init__(self, basech=16, norm=None, activation="relu",
                 has_ltc=True, has_gtc=True, gtc_frozen=False,
                 recurrent_block_type="convgru"):
        super().__init__()
        assert has_ltc or has_gtc
        self.has_ltc = has_ltc
        self.has_gtc = has_gtc
        self.gtc_frozen = gtc_frozen

        if has_ltc:
            self.pred_map = nn.Sequential(
                ConvLayer(2 * basech, basech, 3, 1, 1, activation, norm),
                ConvLayer(basech, 1, 3, 1, 1, "sigmoid", norm),
            )
            self.local_fusion = nn.Sequential(
                ResidualBlock(3 * basech, 3 * basech, norm=norm),
                ConvLayer(3 * basech, basech, 3, 1, 1, None, norm),
            )
        if has_gtc:
            self.lstm = RecurrentConvLayer(basech, basech, 3, 1, 1,
                                           recurrent_block_type, activation, norm)
            self.global_fusion = ConvLayer(2 * basech, basech, 1, 1, 0,
                                           activation, norm)
        self.state = None  # concatenated (fwd|bwd) recurrent state

    def reset_states(self):
        self.state = None

    def detach_states(self):
        """Truncate BPTT: keep state values, drop the autograd history."""
        def _d(s):
            if s is None:
                return None
            if isinstance(s, tuple):
                return tuple(_d(v) for v in s)
            return s.detach()
        self.state = _d(self.state)

    def _local_batched(self, x, N):
        """All N triplets in one batched call set (same math as the
        reference's per-frame loop, ESR:models/model.py:133-144, with the
        same edge duplication).  x is FRAME-major [N*B, C, H, W]: frame
        shifts are contiguous slice concatenations (no index tensors, no 5D
        round trips)."""
        B = x.size(0) // N
        f0 = torch.cat([x[:B], x[:-B]], dim=0)       # frame i-1 (dup frame 0)
        f2 = torch.cat([x[B:], x[-B:]], dim=0)       # frame i+1 (dup frame N-1)
        pairs = torch.cat([torch.cat([f0, x], dim=1),
                           torch.cat([x, f2], dim=1)], dim=0)
        maps = self.pred_map(pairs)
        m0, m1 = maps[: N * B], maps[N * B:]
        out = self.local_fusion(torch.cat([f0 * m0, x, f2 * m1], dim=1))
        return out + x

    def _global(self, feats, N):
        """Bi-directional shared-weight GRU scan over frame-major input;
        fwd and bwd direction run in one batched cell call per step."""
        B = feats.size(0) // N
        state = None if self.gtc_frozen else self.state
        outs = []
        for n in range(N):
            fwd_in = feats[n * B:(n + 1) * B]
            bwd_in = feats[(N - 1 - n) * B:(N - n) * B]
            out, state = self.lstm(torch.cat([fwd_in, bwd_in], dim=0), state)
            if self.gtc_frozen:
                state = None
            outs.append(out)                          # [2B, C, H, W]
        self.state = None if self.gtc_frozen else state

        fused = torch.cat(
            [torch.cat([outs[n][:B], outs[N - 1 - n][B:]], dim=1)
             for n in range(N)], dim=0)               # [N*B, 2C, H, W]
        return self.global_fusion(fused)

    def forward_frames(self, x, N):
        """Frame-major entry: x [N*B, C, H, W]."""
        feats = self._local_batched(x, N) if self.has_ltc else x
        if self.has_gtc:
            feats = self._global(feats, N)
        return feats + x

    def forward(self, x):
        """5D-compat entry: x [B, N, C, H, W]."""
        B, N, C, H, W = x.shape
        xf = x.transpose(0, 1).reshape(N * B, C, H, W)
        out = self.forward_frames(xf, N)
        return out.reshape(N, B, C, H, W).transpose(0, 1)


class STFusion(nn.Module):
    """Deformable alignment of each frame to the centre frame, spatial +
    channel attention fusion, and 3-level decoder
    (parity: ESR:models/model.py:156-291)."""

    def __init__(self, basech=16, num_frame=3, norm=None, activation="relu",
                 has_dcnatten=True, has_scaleaggre=True, upsampler="bilinear",
                 deformable_groups=8):
        super().__init__()
        assert has_dcnatten or has_scaleaggre
        assert num_frame >= 3 and (num_frame + 1) % 2 == 0
        self.has_dcnatten = has_dcnatten
        self.has_scaleaggre = has_scaleaggre
        self.num_frame = num_frame
        self.mid_idx = (num_frame - 1) // 2

        if has_dcnatten:
            self.offset = nn.Sequential(
                ConvLayer(2 * basech, basech, 3, 1, 1, activation, norm),
                ConvLayer(basech, basech, 3, 1, 1, None, norm),
            )
            self.dcn = DeformAlign2d(basech, basech, 3, stride=1, padding=1,
                                     dilation=1, deformable_groups=deformable_groups)
            self.convblock = nn.Sequential(
                ConvLayer(2 * basech, basech, 3, 1, 1, activation, norm),
                ConvLayer(basech, basech, 3, 1, 1, None, norm),
            )
            self.kernel = ConvLayer(basech, 2, 1, 1, 0, "sigmoid", norm)
            self.fc = nn.Sequential(
                MLP(basech, basech // 2, 2 * basech, 2), nn.Sigmoid())
            self.dcn_fusion = nn.Sequential(
                ConvLayer(2 * basech, basech, 3, 1, 1, activation, norm),
                ConvLayer(basech, basech, 3, 1, 1, None, norm),
            )

        self.dense_fusion = nn.Sequential(
            ConvLayer(num_frame * basech, basech, 3, 1, 1, activation, norm),
            ConvLayer(basech, basech, 3, 1, 1, None, norm),
        )

        if has_scaleaggre:
            self.attens = nn.ModuleList([
                ConvLayer(basech, 1, 3, 1, 1, "sigmoid", norm),
                ConvLayer(basech // 2, 1, 3, 1, 1, "sigmoid", norm),
                ConvLayer(basech // 4, 1, 3, 1, 1, "sigmoid", norm),
            ])

        Up = {"bilinear": UpsampleConvLayer,
              "pixelshuffle": PixelShuffleUpsample}[upsampler]
        self.recons = nn.ModuleList([
            Up(basech, basech // 2, 3, 1, 1, norm=norm),
            Up(basech // 2, basech // 4, 3, 1, 1, norm=norm),
            Up(basech // 4, basech // 8, 3, 1, 1, norm=norm),
        ])

    def fuse(self, feat0, feat1):
        """Deformable alignment of feat0 onto feat1 + spatial/channel
        attention fusion (parity: ESR:models/model.py:208-231).  Works on
        any leading batch, so dense_fuse batches all non-centre frames
        through it in ONE call."""
        B, C, H, W = feat0.shape
        offset_feat = self.offset(torch.cat([feat0, feat1], dim=1))
        aligned = F.relu(self.dcn(feat0, offset_feat))
        feat = self.convblock(torch.cat([aligned, feat1], dim=1))
        spatial_k = self.kernel(feat)                                  # [B,2,H,W]
        pooled = feat.reshape(B, C, H * W).transpose(1, 2).max(1, keepdim=True)[0]
        channel_k = self.fc(pooled).transpose(1, 2).unsqueeze(-1)      # [B,2C,1,1]
        y0 = aligned * spatial_k[:, :1] * channel_k[:, :C]
        y1 = feat1 * spatial_k[:, 1:2] * channel_k[:, C:]
        return self.dcn_fusion(torch.cat([y0, y1], dim=1))

    def dense_fuse_frames(self, x, N):
        """x FRAME-major [N*B, C, H, W]; aligns every non-centre frame to
        the centre in one batched DCN/attention pass."""
        B = x.size(0) // N
        m = self.mid_idx
        mid = x[m * B:(m + 1) * B]
        if self.has_dcnatten:
            f0 = torch.cat([x[:m * B], x[(m + 1) * B:]], dim=0)
            f1 = mid.repeat(N - 1, 1, 1, 1)
            fused = self.fuse(f0, f1)                  # [(N-1)*B, C, H, W]
            out = torch.cat([fused[i * B:(i + 1) * B] for i in range(N - 1)]
                            + [mid], dim=1)
        else:
            # channel concat of the N frames per batch item
            out = torch.cat([x[i * B:(i + 1) * B] for i in range(N)], dim=1)
        return self.dense_fusion(out)

    def scale_aggre(self, x, feats, N, idx):
        """feats FRAME-major [N*B, C, h, w]."""
        if self.has_scaleaggre:
            B = feats.size(0) // N
            flat = feats * self.attens[idx](feats)
            x = x + flat.view(N, B, *flat.shape[1:]).mean(0)  # free view
        return self.recons[idx](x)

    def forward_frames(self, x, feats_list, N):
        assert N == self.num_frame
        out = self.dense_fuse_frames(x, N)
        for idx, feats in enumerate(feats_list):
            out = self.scale_aggre(out, feats, N, idx)
        return out

    def forward(self, x, feats_list):
        """5D-compat entry: x [B, N, C, H, W]; feats_list entries
        [B*N, c, h, w] batch-major (as the encoder produces for a
        batch-major flatten)."""
        B, N = x.shape[:2]
        xf = x.transpose(0, 1).reshape(N * B, *x.shape[2:])
        ff = [f.view(B, N, *f.shape[1:]).transpose(0, 1)
              .reshape(N * B, *f.shape[1:]) for f in feats_list]
        return self.forward_frames(xf, ff, N)


@register_model("ESRNet")
@register_model("DeepRecurrNet")  # reference-config compatibility alias
class ESRNet(nn.Module):
    """Recurrent event-stream SR network (reference: DeepRecurrNet,
    ESR:models/model.py:294-344).

    Input: [B, N, inch, kH, kW] scaled count maps (LR events splatted on the
    HR grid); output: [B, inch, kH, kW] predicted HR count map for the
    middle frame.
    """

    DOWN_SCALE = 8

    def __init__(self, inch=2, basech=16, num_frame=3, norm=None,
                 activation="relu", has_ltc=True, has_gtc=True,
                 gtc_frozen=False, has_dcnatten=True, has_scaleaggre=True,
                 upsampler="bilinear", recurrent_block_type="convgru",
                 deformable_groups=8):
        super().__init__()
        d = self.DOWN_SCALE
        self.head = ConvLayer(inch, basech, 3, 1, 1, activation, norm)
        self.feat_extract = FeatsExtract(basech, norm, activation)
        self.time_propagate = TimePropagation(
            d * basech, norm, activation, has_ltc, has_gtc, gtc_frozen,
            recurrent_block_type)
        self.spacetime_fuse = STFusion(
            d * basech, num_frame, norm, activation, has_dcnatten,
            has_scaleaggre, upsampler, deformable_groups)
        self.tail = ConvLayer(basech, inch, 3, 1, 1, "relu", norm)

    def reset_states(self):
        self.time_propagate.reset_states()

    def detach_states(self):
        self.time_propagate.detach_states()

    @property
    def num_parameters(self):
        return sum(p.numel() for p in self.parameters() if p.requires_grad)

    def forward(self, x):
        B, N, C, H, W = x.shape
        x, box = _pad_multiple(x, self.DOWN_SCALE)
        # frame-major layout [N*B, C, H, W]: every temporal op below is a
        # contiguous batch-slice concatenation — no 5D round trips, no
        # index tensors (hipGraph-capture-safe), channels_last-compatible
        x = x.transpose(0, 1).reshape(N * B, C, x.size(-2), x.size(-1))
        x = self.head(x)
        feats_list = self.feat_extract(x)
        deep = self.time_propagate.forward_frames(feats_list[0], N)
        out = self.spacetime_fuse.forward_frames(deep, feats_list, N)
        out = self.tail(out)
        return _crop(out, box)

    def forward_sequence(self, frames, seqn: int):
        """BPTT over all sliding seqn-windows of a sequence with the
        head/encoder computed ONCE per unique frame.

        frames: [B, L, C, H, W] (L >= seqn).  Returns the list of
        L-seqn+1 outputs [B, inch, H, W] — mathematically identical to
        calling forward() per window (the reference's loop,
        ESR:train_ours_cnt_seq.py:217-232): per-frame encoders are
        frame-local, so overlapping windows share their features; autograd
        accumulates every window's gradient into the shared encoder pass.
        In frame-major layout each window's features are a contiguous
        slice — no gather, no copy.
        """
        B, L, C, H, W = frames.shape
        nW = L - seqn + 1
        x, box = _pad_multiple(frames, self.DOWN_SCALE)
        x = x.transpose(0, 1).reshape(L * B, C, x.size(-2), x.size(-1))
        x = self.head(x)
        feats_list = self.feat_extract(x)

        # temporal propagation is stateful -> sequential over windows
        deeps = [self.time_propagate.forward_frames(
            feats_list[0][w * B:(w + seqn) * B], seqn) for w in range(nW)]

        # spatio-temporal fusion + decoder have NO cross-window state:
        # run ALL windows in one batched pass (combined batch nW*B,
        # frame-major: frame n block = that frame of every window)
        def window_major(per_window, n):
            return torch.cat([d[n * B:(n + 1) * B] for d in per_window], dim=0)

        deep_all = torch.cat([window_major(deeps, n) for n in range(seqn)],
                             dim=0)
        feats_all = [
            torch.cat([torch.cat([f[(w + n) * B:(w + n + 1) * B]
                                  for w in range(nW)], dim=0)
                       for n in range(seqn)], dim=0)
            for f in feats_list]
        out_all = self.spacetime_fuse.forward_frames(deep_all, feats_all, seqn)
        out_all = _crop(self.tail(out_all), box)
        return [out_all[w * B:(w + 1) * B] for w in range(nW)]
