"""Equivalence tests: the frame-major batched TimePropagation/STFusion must
match a straightforward per-frame-loop computation of the same math
(the reference's loops, ESR:models/model.py:126-153, :233-251)."""

import torch
import torch.nn.functional as F

from esr_amd.models.esrnet import STFusion, TimePropagation


def _loop_ltc(tp, x):
    B, N, C, H, W = x.shape
    feats = []
    for i in range(N):
        i0, i1, i2 = (0, 0, 1) if i == 0 else \
            ((N - 2, N - 1, N - 1) if i == N - 1 else (i - 1, i, i + 1))
        f0, f1, f2 = x[:, i0], x[:, i1], x[:, i2]
        m0 = tp.pred_map(torch.cat([f0, f1], dim=1))
        m1 = tp.pred_map(torch.cat([f1, f2], dim=1))
        out = tp.local_fusion(torch.cat([f0 * m0, f1, f2 * m1], dim=1))
        feats.append(out + f1)
    return torch.stack(feats, dim=1)


def _loop_gtc(tp, feats):
    B, N, C, H, W = feats.shape
    rev_idx = list(reversed(range(N)))
    rev = feats[:, rev_idx]
    state = state_rev = None
    xs, rs = [], []
    for i in range(N):
        x, state = tp.lstm(feats[:, i], state)
        r, state_rev = tp.lstm(rev[:, i], state_rev)
        xs.append(x)
        rs.append(r)
    x = torch.stack(xs, 1)
    r = torch.stack(rs, 1)[:, rev_idx]
    fused = torch.cat([x, r], dim=2).reshape(B * N, -1, H, W)
    fused = tp.global_fusion(fused)
    return fused.reshape(B, N, C, H, W)


def test_time_propagation_matches_loop():
    torch.manual_seed(0)
    tp = TimePropagation(basech=8)
    x = torch.randn(2, 3, 8, 8, 8)
    tp.reset_states()
    out = tp(x)
    ref = _loop_gtc(tp, _loop_ltc(tp, x)) + x
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max().item()


def test_time_propagation_state_across_calls_matches_loop():
    torch.manual_seed(1)
    tp = TimePropagation(basech=4, has_ltc=False)
    x1 = torch.randn(1, 3, 4, 8, 8)
    x2 = torch.randn(1, 3, 4, 8, 8)
    tp.reset_states()
    tp(x1)
    out2 = tp(x2)

    # loop reference with explicit fwd/bwd states carried across calls
    state = state_rev = None

    def gtc(feats):
        nonlocal state, state_rev
        B, N, C, H, W = feats.shape
        rev_idx = list(reversed(range(N)))
        rev = feats[:, rev_idx]
        xs, rs = [], []
        for i in range(N):
            x, state_ = tp.lstm(feats[:, i], state)
            r, state_rev_ = tp.lstm(rev[:, i], state_rev)
            state, state_rev = state_, state_rev_
            xs.append(x)
            rs.append(r)
        x = torch.stack(xs, 1)
        r = torch.stack(rs, 1)[:, rev_idx]
        fused = torch.cat([x, r], dim=2).reshape(B * N, -1, H, W)
        return tp.global_fusion(fused).reshape(B, N, C, H, W)

    gtc(x1)
    ref2 = gtc(x2) + x2
    assert torch.allclose(out2, ref2, atol=1e-5)


def test_stfusion_matches_loop():
    torch.manual_seed(2)
    sf = STFusion(basech=8, num_frame=3)
    x = torch.randn(2, 3, 8, 8, 8)
    feats_list = [torch.randn(6, 8, 8, 8), torch.randn(6, 4, 16, 16),
                  torch.randn(6, 2, 32, 32)]
    out = sf(x, feats_list)

    # per-frame loop reference
    mid = x[:, 1]
    fused = [sf.fuse(x[:, i], mid) for i in (0, 2)]
    df = sf.dense_fusion(torch.cat(fused + [mid], dim=1))
    cur = df
    B, N = 2, 3
    for idx, feats in enumerate(feats_list):
        f = feats.view(B, N, -1, feats.size(-2), feats.size(-1))
        flat = f.reshape(B * N, *f.shape[2:])
        flat = flat * sf.attens[idx](flat)
        cur = cur + flat.view_as(f).mean(1)
        cur = sf.recons[idx](cur)
    assert torch.allclose(out, cur, atol=1e-5), (out - cur).abs().max().item()
