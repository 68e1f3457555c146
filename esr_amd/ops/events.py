"""Event-stream representation ops.

Vectorized (scatter-based) implementations of the event<->grid conversions the
reference computes with per-event Python loops / Cython kernels
(ESR:dataloader/encodings.py, ESR:dataloader/cython_event_redistribute/
event_redistribute.pyx, ESR:dataloader/cython_cnt2event/cnt2event.pyx).

Every function here runs on CPU and GPU tensors alike.  On a GPU box the
splatting ops dispatch to the hand-written HIP kernels in
``esr_amd.ops.native`` (CDNA4 / gfx950); the torch implementation is the
CPU oracle the HIP kernels are tested against.

Semantics parity notes (vs the reference):
  * Out-of-range events are redirected to pixel (0,0) with weight 0 —
    identical to the reference's masking (ESR:dataloader/encodings.py:36-41).
  * Time-bin edges are right-inclusive, matching the reference's binary-search
    windowing (ESR:dataloader/encodings.py:226-231).  The reference can count
    an event lying exactly on an interior bin edge into *both* neighbouring
    bins (its `beg` search is also inclusive); we deliberately do not
    replicate that double-count (measure-zero for real float timestamps).
"""

from __future__ import annotations

import torch

__all__ = [
    "events_to_image",
    "events_to_channels",
    "events_to_stack_no_polarity",
    "events_to_stack_polarity",
    "events_to_voxel",
    "stack_to_count",
    "redistribute_stack",
    "redistribute_count",
    "event_formatting",
    "normalize_events",
    "scaled_count_encoding",
    "events_to_mask",
    "events_polarity_mask",
    "event_conversion",
    "event_restore",
]


def _mask_and_flatten(xs: torch.Tensor, ys: torch.Tensor, ws: torch.Tensor,
                      sensor_size) -> tuple[torch.Tensor, torch.Tensor]:
    """Clamp out-of-range events to pixel 0 with weight 0; return flat indices
    and weights.  Matches ESR:dataloader/encodings.py:243-268 semantics."""
    H, W = int(sensor_size[0]), int(sensor_size[1])
    xi = xs.long()
    yi = ys.long()
    valid = (xi >= 0) & (xi < W) & (yi >= 0) & (yi < H)
    xi = torch.where(valid, xi, torch.zeros_like(xi))
    yi = torch.where(valid, yi, torch.zeros_like(yi))
    ws = torch.where(valid, ws, torch.zeros_like(ws))
    return yi * W + xi, ws


def events_to_image(xs: torch.Tensor, ys: torch.Tensor, ws: torch.Tensor,
                    sensor_size) -> torch.Tensor:
    """Scatter-accumulate per-event weights into an H x W image.

    Parity: ESR:dataloader/encodings.py:243-268 (`events_to_image`).
    """
    H, W = int(sensor_size[0]), int(sensor_size[1])
    idx, w = _mask_and_flatten(xs, ys, ws.float(), sensor_size)
    img = torch.zeros(H * W, dtype=torch.float32, device=xs.device)
    img.scatter_add_(0, idx, w)
    return img.view(H, W)


def events_to_channels(xs: torch.Tensor, ys: torch.Tensor, ps: torch.Tensor,
                       sensor_size) -> torch.Tensor:
    """Two-channel polarity count map (ch0: positive, ch1: negative) — the
    core representation of the framework.

    Parity: ESR:dataloader/encodings.py:289-304 (`events_to_channels`):
    channel weights are ps*relu(ps) and ps*min(ps,0), i.e. +1 per event of
    the matching polarity when ps in {-1,+1}.
    """
    ps = ps.float()
    pos_w = ps * ps.clamp(min=0)
    neg_w = ps * ps.clamp(max=0)
    pos = events_to_image(xs, ys, pos_w, sensor_size)
    neg = events_to_image(xs, ys, neg_w, sensor_size)
    return torch.stack([pos, neg])


def _time_bins(ts: torch.Tensor, B: int) -> torch.Tensor:
    """Per-event bin index with right-inclusive interior edges.

    Matches the reference's [tstart, tend]-inclusive binary-search windows
    (ESR:dataloader/encodings.py:226-231) minus the boundary double-count.
    """
    if len(ts) == 0:
        return ts.long()
    t0 = ts[0]
    dt = ts[-1] - t0 + 1e-6
    delta = dt / B
    if B == 1:
        return torch.zeros_like(ts, dtype=torch.long)
    edges = t0 + delta * torch.arange(1, B, device=ts.device, dtype=ts.dtype)
    return torch.bucketize(ts.contiguous(), edges, right=False).clamp_(0, B - 1)


def _degenerate(ts: torch.Tensor) -> bool:
    # Reference early-out: all-zero timestamps or <=3 events -> zero tensor
    # (ESR:dataloader/encodings.py:219-220).
    return bool(ts.sum() == 0) or len(ts) <= 3


def events_to_stack_no_polarity(xs, ys, ts, ps, B: int, sensor_size) -> torch.Tensor:
    """Signed event stack, B time bins (B x H x W).

    Parity: ESR:dataloader/encodings.py:204-240.
    """
    H, W = int(sensor_size[0]), int(sensor_size[1])
    B = int(B)
    if _degenerate(ts):
        return torch.zeros(B, H, W, device=xs.device)
    bins = _time_bins(ts, B)
    idx, w = _mask_and_flatten(xs, ys, ps.float(), sensor_size)
    flat = torch.zeros(B * H * W, dtype=torch.float32, device=xs.device)
    flat.scatter_add_(0, bins * (H * W) + idx, w)
    return flat.view(B, H, W)


def events_to_stack_polarity(xs, ys, ts, ps, B: int, sensor_size) -> torch.Tensor:
    """Polarity-split event stack (2 x B x H x W).

    Parity: ESR:dataloader/encodings.py:153-201 (weights ps*relu(ps) /
    ps*min(ps,0) per polarity channel).  The reference's degenerate branch
    returns the wrong rank ([B,H,W]); we return the documented [2,B,H,W].
    """
    H, W = int(sensor_size[0]), int(sensor_size[1])
    B = int(B)
    if _degenerate(ts):
        return torch.zeros(2, B, H, W, device=xs.device)
    bins = _time_bins(ts, B)
    ps = ps.float()
    idx, _ = _mask_and_flatten(xs, ys, ps, sensor_size)
    out = torch.zeros(2, B * H * W, dtype=torch.float32, device=xs.device)
    lin = bins * (H * W) + idx
    _, pos_w = _mask_and_flatten(xs, ys, ps * ps.clamp(min=0), sensor_size)
    _, neg_w = _mask_and_flatten(xs, ys, ps * ps.clamp(max=0), sensor_size)
    out[0].scatter_add_(0, lin, pos_w)
    out[1].scatter_add_(0, lin, neg_w)
    return out.view(2, B, H, W)


def events_to_voxel(xs, ys, ts, ps, num_bins: int, sensor_size) -> torch.Tensor:
    """Temporal-bilinear voxel grid from events with ts normalized to [0,1].

    Parity: ESR:dataloader/encodings.py:271-286 (`events_to_voxel`): per bin
    weight = max(0, 1-|ts*(B-1) - bi|).  Each event contributes to at most two
    adjacent bins; implemented as two scatters.
    """
    H, W = int(sensor_size[0]), int(sensor_size[1])
    B = int(num_bins)
    idx, w = _mask_and_flatten(xs, ys, ps.float(), sensor_size)
    tn = (ts.float() * (B - 1)) if B > 1 else torch.zeros_like(ts, dtype=torch.float32)
    lo = tn.floor().clamp(0, B - 1)
    frac = tn - lo
    lo_i = lo.long()
    hi_i = (lo_i + 1).clamp(max=B - 1)
    flat = torch.zeros(B * H * W, dtype=torch.float32, device=xs.device)
    flat.scatter_add_(0, lo_i * (H * W) + idx, w * (1.0 - frac))
    # hi-bin weight is max(0, 1-|tn - (lo+1)|) = frac for in-range events;
    # when hi_i is clamped (tn == B-1 exactly) frac == 0 so no double count.
    flat.scatter_add_(0, hi_i * (H * W) + idx, w * frac)
    return flat.view(B, H, W)


def stack_to_count(stack: torch.Tensor) -> torch.Tensor:
    """B x TB x H x W signed stack -> B x 2 x H x W polarity count map.

    Parity: ESR:dataloader/encodings.py:652-670 (`stack2cnt`).
    """
    s = stack.detach().round()
    pos = s.clamp(min=0).sum(dim=1)
    neg = (-s.clamp(max=0)).sum(dim=1)
    return torch.stack([pos, neg], dim=1)


def _expand_linspace(counts: torch.Tensor, t0: torch.Tensor, t1: torch.Tensor,
                     mode: str, generator=None) -> torch.Tensor:
    """For each cell i with counts[i] events, produce timestamps spread over
    [t0[i], t1[i]]: linspace (n>1; t0 for n==1) or uniform random."""
    total = int(counts.sum().item())
    device = counts.device
    if total == 0:
        return torch.zeros(0, device=device)
    reps = counts
    starts = torch.cumsum(reps, 0) - reps           # first flat index of each cell
    cell_of = torch.repeat_interleave(torch.arange(len(reps), device=device), reps)
    j = torch.arange(total, device=device) - starts[cell_of]
    n = reps[cell_of].float()
    t0e = t0[cell_of]
    t1e = t1[cell_of]
    if mode == "linear":
        denom = (n - 1).clamp(min=1)
        t = t0e + (t1e - t0e) * (j.float() / denom)
    elif mode == "random":
        r = torch.rand(total, device=device, generator=generator)
        t = t0e + (t1e - t0e) * r
    else:
        raise ValueError(f"unsupported redistribute mode: {mode}")
    return t


def redistribute_stack(stack: torch.Tensor, mode: str = "linear",
                       generator=None, capacity: int | None = None
                       ) -> torch.Tensor:
    """Count stack -> event cloud, the inverse of the splatting ops.

    Input: [B, C, Y, X] (no-polarity) or [B, P, C, Y, X] (polarity) count
    stack.  Output: [B, N_max, 4] float events (x, y, t, p) sorted by t and
    zero-padded per batch item.

    Parity: ESR:dataloader/cython_event_redistribute/event_redistribute.pyx:17-154
    and ESR:dataloader/encodings.py:366-463: a cell with value v at bin c
    emits |v| events at that pixel with timestamps in
    (c/C + 1/(100C), (c+1)/C], polarity sign(v); per-item global sort by t.

    On GPU with the native extension this runs as one device pipeline
    (prefix scan + scatter + segmented radix sort, redistribute.hip) with
    no host syncs when ``capacity`` (max events per item) is given —
    graph-capturable; without ``capacity`` one sync computes the true max
    length.  The vectorized torch path below is the CPU oracle.
    """
    native = _native_redistribute(stack, mode, generator, capacity)
    if native is not None:
        return native
    if stack.dim() == 5:
        Bb, P, C, Y, X = stack.shape
        s = stack.round().reshape(Bb, P * C, Y, X)
        bins_of_channel = torch.arange(P * C, device=stack.device) % C
    elif stack.dim() == 4:
        Bb, C, Y, X = stack.shape
        s = stack.round()
        bins_of_channel = torch.arange(C, device=stack.device)
    else:
        raise ValueError("stack must be 4D [B,C,Y,X] or 5D [B,P,C,Y,X]")

    device = stack.device
    num_bins = C
    per_item = []
    for b in range(Bb):
        entry = s[b]                                    # [C', Y, X]
        nz = entry.nonzero(as_tuple=False)              # row-major (c', y, x)
        if nz.numel() == 0:
            per_item.append(torch.zeros(1, 4, device=device))
            continue
        vals = entry[nz[:, 0], nz[:, 1], nz[:, 2]]
        counts = vals.abs().long()
        c = bins_of_channel[nz[:, 0]].float()
        t0 = c / num_bins + 1.0 / (100.0 * num_bins)
        t1 = (c + 1.0) / num_bins
        ts = _expand_linspace(counts, t0, t1, mode, generator)
        cell_of = torch.repeat_interleave(torch.arange(len(counts), device=device), counts)
        ev = torch.empty(len(ts), 4, device=device)
        ev[:, 0] = nz[cell_of, 2].float()               # x
        ev[:, 1] = nz[cell_of, 1].float()               # y
        ev[:, 2] = ts
        ev[:, 3] = torch.where(vals[cell_of] > 0,
                               torch.ones_like(ts), -torch.ones_like(ts))
        order = torch.argsort(ev[:, 2], stable=True)
        per_item.append(ev[order])

    maxlen = max(e.size(0) for e in per_item) if capacity is None \
        else int(capacity)
    out = torch.zeros(Bb, maxlen, 4, device=device)
    for b, e in enumerate(per_item):
        n = min(e.size(0), maxlen)   # fixed capacity: truncate like the
        out[b, :n] = e[:n]           # HIP pipeline (cell-order prefix)
    return out


def _native_redistribute(stack, mode, generator, capacity):
    """Device pipeline dispatch; returns None when the torch path applies."""
    if not stack.is_cuda or mode not in ("linear", "random"):
        return None
    from .native import get_ext
    ext = get_ext()
    if ext is None or not hasattr(ext, "redistribute_stack_hip"):
        return None
    if stack.dim() == 5:
        Bb, P, C, Y, X = stack.shape
        flat = stack.reshape(Bb, P * C, Y, X)
    elif stack.dim() == 4:
        C = stack.size(1)
        flat = stack
    else:
        raise ValueError("stack must be 4D [B,C,Y,X] or 5D [B,P,C,Y,X]")
    flat = flat.float().contiguous()
    sync_trim = capacity is None
    if sync_trim:
        # one sync for the true max length (the Cython parity shape);
        # pass capacity explicitly for the graph-capturable path
        capacity = max(1, int(flat.round().abs().sum(dim=(1, 2, 3)).max()))
    seed = 123 if generator is None else \
        int(generator.initial_seed()) & 0xFFFFFFFF
    events, lengths = ext.redistribute_stack_hip(
        flat, C, int(capacity), 0 if mode == "linear" else 1, seed)
    return events


def redistribute_count(cnt: torch.Tensor, mode: str = "linear",
                       generator=None, capacity: int | None = None
                       ) -> torch.Tensor:
    """2-channel count map [B,2,H,W] -> event cloud [B,N,4].

    Parity: ESR:dataloader/cython_cnt2event/cnt2event.pyx:18-116.  A count map
    is the 1-bin case of a polarity stack with ch1 counted as negative events.
    """
    if cnt.dim() != 4 or cnt.size(1) != 2:
        raise ValueError("cnt must be [B,2,H,W]")
    signed = torch.stack([cnt[:, 0], -cnt[:, 1]], dim=1)   # [B,2,H,W]
    return redistribute_stack(signed.unsqueeze(2), mode=mode,
                              generator=generator, capacity=capacity)


def event_formatting(events) -> torch.Tensor:
    """np [4,N] (x,y,t,p) -> float32 torch with t normalized to [0,1].

    Parity: ESR:dataloader/base_dataset.py:26-33.
    """
    ev = torch.as_tensor(events, dtype=torch.float32).clone()
    if ev.numel() and ev.size(1) > 0:
        t = ev[2]
        tmin = t.min()
        denom = (t.max() - tmin)
        ev[2] = (t - tmin) / (denom + 1e-6) if denom > 0 else torch.zeros_like(t)
    return ev


def normalize_events(events: torch.Tensor, sensor_resolution) -> torch.Tensor:
    """[4,N] events -> coordinates normalized by the sensor resolution.

    Parity: ESR:dataloader/h5dataset.py:508-518.
    """
    xs, ys, ts, ps = events[0], events[1], events[2], events[3]
    H, W = sensor_resolution
    return torch.stack([xs / W, ys / H, ts, ps]).float()


def scaled_count_encoding(normalized_events: torch.Tensor, sensor_resolution,
                          mode: str = "cnt", time_bins: int = 1) -> torch.Tensor:
    """Splat normalized events onto an arbitrary pixel grid — the op that
    produces the model input (LR events re-splatted on the HR grid).

    Parity: ESR:dataloader/h5dataset.py:520-536 (`create_scaled_encoding`).
    """
    xs, ys, ts, ps = normalized_events
    H, W = sensor_resolution
    if mode == "cnt":
        return events_to_channels(xs * W, ys * H, ps, sensor_resolution)
    if mode == "stack":
        return events_to_stack_no_polarity(xs * W, ys * H, ts, ps, time_bins,
                                           sensor_resolution)
    if mode == "events":
        return torch.stack([(xs * W).long().float(), (ys * H).long().float(), ts, ps])
    raise ValueError(f"unsupported mode {mode}")


def events_to_mask(xs: torch.Tensor, ys: torch.Tensor, ps: torch.Tensor,
                   sensor_size) -> torch.Tensor:
    """Binary activity mask: |p| written (not accumulated) per pixel.

    Parity: ESR:dataloader/encodings.py:307-331.
    """
    H, W = int(sensor_size[0]), int(sensor_size[1])
    idx, w = _mask_and_flatten(xs, ys, ps.float().abs(), sensor_size)
    mask = torch.zeros(H * W, dtype=torch.float32, device=xs.device)
    mask.scatter_(0, idx, w)
    return mask.view(H, W)


def events_polarity_mask(ps: torch.Tensor) -> torch.Tensor:
    """[N] polarities -> [N, 2] one-hot-ish polarity mask (ch0 pos, ch1
    |neg|).  Parity: ESR:dataloader/encodings.py:334-345."""
    pos = ps.clamp(min=0)
    neg = -ps.clamp(max=0)
    return torch.stack([pos, neg], dim=1)


def event_conversion(event_list: torch.Tensor, time_bins: int, resolution,
                     time_bins_voxel: int | None = None) -> dict:
    """Batched padded event cloud [B, N, 4] (x, y, t, p) -> dict of count /
    voxel / stack encodings.  Parity: ESR:dataloader/encodings.py:536-577
    (vectorized per item; events are re-sorted by t)."""
    if time_bins_voxel is None:
        time_bins_voxel = time_bins
    e_cnt, e_voxel, e_stack = [], [], []
    for entry in event_list.detach():
        order = torch.argsort(entry[:, 2], stable=True)
        ev = entry[order]
        xs, ys, ts, ps = ev[:, 0], ev[:, 1], ev[:, 2], ev[:, 3]
        e_cnt.append(events_to_channels(xs, ys, ps, resolution))
        e_voxel.append(events_to_voxel(xs, ys, ts, ps, time_bins_voxel,
                                       resolution))
        e_stack.append(events_to_stack_no_polarity(xs, ys, ts, ps, time_bins,
                                                   resolution))
    return {"e_cnt": torch.stack(e_cnt), "e_voxel": torch.stack(e_voxel),
            "e_stack": torch.stack(e_stack)}


def event_restore(events: torch.Tensor, resolution) -> torch.Tensor:
    """Normalized event cloud [B, N, 4] -> pixel coordinates with polarity
    snapped to {-1, +1}.  Parity: ESR:dataloader/encodings.py:580-601."""
    ev = events.detach().clone()
    ev[:, :, 0] *= resolution[1]
    ev[:, :, 1] *= resolution[0]
    ev[:, :, 3] = torch.sign(ev[:, :, 3])
    return ev
