"""Event SR dataset: scale-pair resolution, stream windowing, GT alignment.

Behavioural parity with the reference's ``H5Dataset``
(ESR:dataloader/h5dataset.py:21-726) on top of the EVS store:

  * input/GT resolution pairs from ori_scale x scale
    (ESR:dataloader/h5dataset.py:31-145 — the if-chain generalizes to
    level arithmetic over {ori,down2,down4,down8,down16});
  * 'events' / 'time' / 'frame' windowing (ESR:...:163-252);
  * GT window alignment by count: scale^2 * N events starting at the time
    of the window's first input event (ESR:...:451-475);
  * flip/polarity augmentation keyed on a shared per-sequence seed
    (ESR:...:652-685), event-noise injection (ESR:...:716-726), stream
    pauses (zero input, frozen index — driven by SequenceDataset);
  * the per-item dict with the same keys (ESR:...:374-406).

All encodings use the vectorized ops in esr_amd.ops.events.
"""

from __future__ import annotations

import random

import numpy as np
import torch
import torch.nn.functional as F
from torch.utils.data import Dataset

from ..ops import events as E
from .store import EventStore, GROUP_LEVELS

__all__ = ["EventSRDataset", "resolve_scale_pair"]

_LEVEL_NAMES = {1: "ori", 2: "down2", 4: "down4", 8: "down8", 16: "down16"}


def resolve_scale_pair(sensor_resolution, ori_scale: str, scale: int,
                       need_gt_events: bool, real_world_test: bool = False):
    """Compute (inp_prefix, inp_res, gt_prefix, gt_res, inp_down_res).

    Parity: ESR:dataloader/h5dataset.py:31-145.  `ori_scale` names the group
    the *input* events come from; `scale` is the SR factor; the GT group is
    the one `scale`x finer than the input group.
    """
    H, W = sensor_resolution
    if real_world_test:
        if ori_scale != "down8" or need_gt_events:
            raise ValueError("real_world_test supports ori_scale=down8 "
                             "without GT events (ESR:h5dataset.py:44-59)")
        inp_prefix = "down8_real"
        inp_res = [round(H / 8), round(W / 8)]
        gt_prefix = inp_prefix
        lvl = 8 // scale if scale in (2, 4, 8) else 1
        gt_res = [round(H / lvl), round(W / lvl)] if scale in (2, 4, 8) else [H, W]
        down_res = [round(r / scale) for r in inp_res]
        return inp_prefix, inp_res, gt_prefix, gt_res, down_res

    if ori_scale not in GROUP_LEVELS:
        raise ValueError(f"unknown ori_scale {ori_scale}")
    level = GROUP_LEVELS[ori_scale]
    inp_res = [round(H / level), round(W / level)]
    inp_down_res = [round(r / scale) for r in inp_res]
    if not need_gt_events:
        return ori_scale, inp_res, ori_scale, [r * scale for r in inp_res], inp_down_res
    if scale == 1:
        return ori_scale, inp_res, ori_scale, list(inp_res), inp_down_res
    gt_level = level // scale
    if gt_level * scale != level or gt_level not in _LEVEL_NAMES:
        raise ValueError(
            f"scale {scale} has no GT group for ori_scale {ori_scale}")
    gt_prefix = _LEVEL_NAMES[gt_level]
    gt_res = [round(H / gt_level), round(W / gt_level)]
    return ori_scale, inp_res, gt_prefix, gt_res, inp_down_res


class EventSRDataset(Dataset):
    """One EVS sequence, windowed into SR training items."""

    def __init__(self, path, config: dict):
        super().__init__()
        self.config = config
        self.store = EventStore(path)
        self.sensor_resolution = self.store.sensor_resolution

        self.scale = config["scale"]
        self.ori_scale = config["ori_scale"]
        self.time_bins = int(config["time_bins"])
        self.need_gt_events = config.get("need_gt_events", False)
        self.need_gt_frame = config.get("need_gt_frame", False)
        self.real_world_test = config.get("real_world_test", False)
        self.custom_resolution = config.get("custom_resolution", None)
        self.dataset_length = config.get("dataset_length", None)
        self.add_noise = config.get("add_noise", {"enabled": False})
        self.augment_cfg = config.get("data_augment", {"enabled": False})
        self.hot_filter = config.get("hot_filter", {"enabled": False})
        # selective item fields: the flagship trainer consumes 3 of the 21
        # item-dict entries; listing them here skips the unused encodings
        # and interpolations in the CPU workers (~3x items/s at the
        # flagship shape — the 8-GPU input-feed budget, docs/SCALING.md).
        # None (default) builds everything (reference item-dict parity,
        # ESR:dataloader/h5dataset.py:374-406).
        f = config.get("fields", None)
        self.fields = None if f is None else set(f)

        (self.inp_prefix, self.inp_sensor_resolution, self.gt_prefix,
         self.gt_sensor_resolution, self.inp_down_sensor_resolution) = \
            resolve_scale_pair(self.sensor_resolution, self.ori_scale,
                               self.scale, self.need_gt_events,
                               self.real_world_test)

        # metadata (parity: ESR:dataloader/h5dataset.py:147-161)
        self.num_events = self.store.num_events(self.inp_prefix)
        self.num_gt_events = (self.store.num_events(self.gt_prefix)
                              if self.need_gt_events else None)
        ts = self.store.ts(self.inp_prefix)
        self.t0 = float(ts[0]) if len(ts) else 0.0
        self.tk = float(ts[-1]) if len(ts) else 0.0
        self.duration = self.tk - self.t0
        self.hot_events = torch.zeros(self.inp_sensor_resolution)
        self.hot_idx = 0
        if self.need_gt_frame:
            self.gt_frame_ts = self.store.image_ts().tolist() \
                if self.store.num_images else []

        self._set_data_mode()

    # ---------------- windowing ----------------

    def _set_data_mode(self):
        cfg = self.config
        self.data_mode = cfg["mode"]
        self.window = cfg["window"]
        self.sliding_window = cfg["sliding_window"]

        if self.data_mode == "events":
            max_length = max(int(self.num_events / (self.window - self.sliding_window)), 0)
            self.length = min(self.dataset_length or max_length, max_length)
            self._compute_k_indices()
        elif self.data_mode == "time":
            max_length = max(int(self.duration / (self.window - self.sliding_window)), 0)
            self.length = min(self.dataset_length or max_length, max_length)
            self._compute_timeblock_indices()
        elif self.data_mode == "frame":
            max_length = self.store.num_images - 1
            self.length = min(self.dataset_length or max_length, max_length)
            self._compute_frame_indices()
        else:
            raise ValueError(f"invalid data mode {self.data_mode}")
        if self.length <= 0:
            raise ValueError("windowing parameters lead to zero-length dataset")

    def _find_ts_index(self, timestamp) -> int:
        ts = self.store.ts(self.inp_prefix)
        idx = int(np.searchsorted(ts, timestamp, side="left"))
        return min(idx, self.num_events - 1)

    def _gt_indices_by_count(self, idx0, idx1):
        """GT window = scale^2 * N events from the time of input event idx0
        (parity: ESR:dataloader/h5dataset.py:451-475)."""
        num_gt = self.scale ** 2 * (idx1 - idx0)
        t0 = float(self.store.ts(self.inp_prefix)[idx0])
        gt_ts = self.store.ts(self.gt_prefix)
        g0 = int(np.searchsorted(gt_ts, t0, side="left"))
        g1 = g0 + num_gt
        if g1 > self.num_gt_events - 1:
            g1 = self.num_gt_events - 1
            g0 = max(g1 - num_gt, 0)
        return g0, g1

    def _compute_k_indices(self):
        self.event_indices, self.gt_event_indices = [], []
        step = self.window - self.sliding_window
        for i in range(self.length):
            idx0 = step * i
            idx1 = min(idx0 + self.window, self.num_events - 1)
            if self.need_gt_events:
                self.gt_event_indices.append(self._gt_indices_by_count(idx0, idx1))
            self.event_indices.append((idx0, idx1))

    def _compute_timeblock_indices(self):
        self.event_indices, self.gt_event_indices = [], []
        start_idx = 0
        for i in range(self.length):
            start_time = (self.window - self.sliding_window) * i + self.t0
            end_idx = self._find_ts_index(start_time + self.window)
            if self.need_gt_events:
                self.gt_event_indices.append(
                    self._gt_indices_by_count(start_idx, end_idx))
            self.event_indices.append((start_idx, end_idx))
            start_idx = end_idx

    def _compute_frame_indices(self):
        self.event_indices, self.gt_event_indices = [], []
        start_idx = 0
        for ts in self.store.image_ts()[: self.length]:
            end_idx = self._find_ts_index(float(ts))
            if self.need_gt_events:
                self.gt_event_indices.append(
                    self._gt_indices_by_count(start_idx, end_idx))
            self.event_indices.append((start_idx, end_idx))
            start_idx = end_idx

    def __len__(self):
        return self.length

    # ---------------- augmentation & noise ----------------

    def _augment_events(self, ev: np.ndarray, resolution, seed: int) -> np.ndarray:
        """Flip/polarity augmentation with per-mechanism seeds (parity:
        ESR:dataloader/h5dataset.py:652-670)."""
        cfg = self.augment_cfg
        xs, ys, ts, ps = ev[0], ev[1], ev[2], ev[3]
        for i, mechanism in enumerate(cfg["augment"]):
            prob = cfg["augment_prob"][i]
            if mechanism == "Horizontal":
                if random.Random(seed).random() < prob:
                    xs = resolution[1] - 1 - xs
            elif mechanism == "Vertical":
                if random.Random(seed + 1).random() < prob:
                    ys = resolution[0] - 1 - ys
            elif mechanism == "Polarity":
                if random.Random(seed + 2).random() < prob:
                    ps = ps * -1
        return np.stack([xs, ys, ts, ps])

    def _augment_frame(self, img: np.ndarray, seed: int) -> np.ndarray:
        cfg = self.augment_cfg
        for i, mechanism in enumerate(cfg["augment"]):
            prob = cfg["augment_prob"][i]
            if mechanism == "Horizontal" and random.Random(seed).random() < prob:
                img = np.flip(img, 1)
            elif mechanism == "Vertical" and random.Random(seed + 1).random() < prob:
                img = np.flip(img, 0)
        return img

    @staticmethod
    def _noise_events(window, sensor_size, seed, noise_level=0.01) -> torch.Tensor:
        """Uniform random noise events appended at t=1 (parity:
        ESR:dataloader/h5dataset.py:715-726)."""
        g = torch.Generator().manual_seed(seed)
        n = int(window * noise_level)
        r = torch.rand(4, n, generator=g)
        x = (r[0] * sensor_size[1]).int()
        y = (r[1] * sensor_size[0]).int()
        t = torch.ones_like(y)
        p = (r[3] * 2).int() * 2 - 1
        return torch.stack([x, y, t, p]).float()

    # ---------------- item assembly ----------------

    def _interp(self, x: torch.Tensor, size, mode) -> torch.Tensor:
        kwargs = {} if mode == "nearest" else {"align_corners": False}
        return F.interpolate(x.unsqueeze(0), size=size, mode=mode, **kwargs).squeeze(0)

    def __getitem__(self, index, pause: bool = False, seed: int | None = None):
        if seed is None:
            seed = random.randint(0, 2 ** 32)
        idx0, idx1 = self.event_indices[index]
        augment = self.augment_cfg.get("enabled", False)

        inp_ev = self.store.events(self.inp_prefix, idx0, idx1)
        if augment:
            inp_ev = self._augment_events(inp_ev, self.inp_sensor_resolution, seed)
        inp = E.event_formatting(inp_ev)

        if self.need_gt_events:
            g0, g1 = self.gt_event_indices[index]
            gt_ev = self.store.events(self.gt_prefix, g0, g1)
            if augment:
                gt_ev = self._augment_events(gt_ev, self.gt_sensor_resolution, seed)
            gt = E.event_formatting(gt_ev)
        else:
            gt = torch.zeros(4, 1)

        if self.add_noise.get("enabled", False):
            noise = self._noise_events(self.window, self.inp_sensor_resolution,
                                       seed, self.add_noise.get("noise_level", 0.01))
            inp = torch.cat([inp, noise], dim=1)

        # GT frames (parity: ESR:dataloader/h5dataset.py:299-315)
        gt_img = torch.zeros([1] + list(self.gt_sensor_resolution))
        gt_img_inp = torch.zeros([1] + list(self.inp_sensor_resolution))
        frame = torch.zeros([1] + list(self.gt_sensor_resolution))
        if self.need_gt_frame and self.store.num_images:
            img = self._get_gt_frame(idx0, idx1)
            if augment:
                img = self._augment_frame(img, seed)
            t = torch.from_numpy(np.ascontiguousarray(img).copy()).float() / 255.0
            t = t.unsqueeze(0)
            gt_img = self._interp(t, self.gt_sensor_resolution, "bicubic").clamp(0, 1)
            gt_img_inp = self._interp(t, self.inp_sensor_resolution, "bicubic").clamp(0, 1)
        if self.data_mode == "frame" and self.store.num_images:
            img = self.store.image(index)
            if augment:
                img = self._augment_frame(img, seed)
            t = torch.from_numpy(np.ascontiguousarray(img).copy()).float().unsqueeze(0) / 255.0
            frame = self._interp(t, self.gt_sensor_resolution, "bicubic").clamp(0, 1)

        if pause:  # zero-input item (parity: ESR:dataloader/h5dataset.py:318-319)
            inp = torch.zeros(4, 1)

        inp_res = self.inp_sensor_resolution
        gt_res = self.gt_sensor_resolution

        fields = self.fields
        custom_on = self.custom_resolution is not None

        def need(*keys):
            return fields is None or any(k in fields for k in keys)

        hot_on = self.hot_filter.get("enabled", False)
        need_cnt = hot_on or custom_on or \
            need("inp_cnt", "inp_bicubic_cnt", "inp_near_cnt")
        need_stack = hot_on or \
            need("inp_stack", "inp_bicubic_stack", "inp_near_stack")
        need_norm = custom_on or \
            need("inp_scaled_cnt", "inp_scaled_stack",
                 "inp_down_cnt", "inp_down_scaled_cnt")
        need_down = custom_on or need("inp_down_cnt", "inp_down_scaled_cnt")

        out = {"gt_img": gt_img, "gt_inp_size_img": gt_img_inp,
               "frame": frame}
        inp_cnt = inp_stack = None
        if need_stack:
            inp_stack = E.events_to_stack_no_polarity(
                inp[0], inp[1], inp[2], inp[3], self.time_bins, inp_res)
        if need_cnt:
            inp_cnt = E.events_to_channels(inp[0], inp[1], inp[3], inp_res)
        if hot_on:
            hot_mask = self._hot_mask(inp, inp_res)
            inp_cnt = inp_cnt * hot_mask
            inp_stack = inp_stack * hot_mask
        if inp_cnt is not None:
            out["inp_cnt"] = inp_cnt
        if inp_stack is not None:
            out["inp_stack"] = inp_stack
        if need("inp_bicubic_cnt"):
            out["inp_bicubic_cnt"] = self._interp(inp_cnt, gt_res, "bicubic")
        if need("inp_bicubic_stack"):
            out["inp_bicubic_stack"] = self._interp(inp_stack, gt_res,
                                                    "bicubic")
        if need("inp_near_cnt"):
            out["inp_near_cnt"] = self._interp(inp_cnt, gt_res, "nearest")
        if need("inp_near_stack"):
            out["inp_near_stack"] = self._interp(inp_stack, gt_res, "nearest")

        inp_scaled_cnt = inp_down_cnt = inp_down_scaled_cnt = None
        if need_norm:
            norm_ev = E.normalize_events(inp, inp_res)
            if custom_on or need("inp_scaled_cnt"):
                inp_scaled_cnt = E.scaled_count_encoding(norm_ev, gt_res,
                                                         "cnt")
                out["inp_scaled_cnt"] = inp_scaled_cnt
            if need("inp_scaled_stack"):
                out["inp_scaled_stack"] = E.scaled_count_encoding(
                    norm_ev, gt_res, "stack", self.time_bins)
            if need_down:
                inp_down_cnt, inp_down_scaled_cnt = \
                    self._unsupervised_pair(norm_ev)
                out["inp_down_cnt"] = inp_down_cnt
                out["inp_down_scaled_cnt"] = inp_down_scaled_cnt

        if need("gt_stack"):
            out["gt_stack"] = E.events_to_stack_no_polarity(
                gt[0], gt[1], gt[2], gt[3], self.time_bins, gt_res)
        gt_cnt = None
        if custom_on or need("gt_cnt"):
            gt_cnt = E.events_to_channels(gt[0], gt[1], gt[3], gt_res)
            out["gt_cnt"] = gt_cnt

        if custom_on:
            cr = list(self.custom_resolution)
            cr_up = [c * self.scale for c in cr]
            cr_dn = [round(c / self.scale) for c in cr]
            out["inp_custom_cnt"] = self._interp(inp_cnt, cr,
                                                 "bicubic").round()
            out["inp_custom_scaled_cnt"] = self._interp(
                inp_scaled_cnt, cr_up, "bicubic").round()
            out["inp_custom_down_cnt"] = self._interp(
                inp_down_cnt, cr_dn, "bicubic").round()
            out["inp_custom_down_scaled_cnt"] = self._interp(
                inp_down_scaled_cnt, cr, "bicubic").round()
            out["gt_custom_cnt"] = self._interp(gt_cnt, cr_up,
                                                "bicubic").round()
        elif fields is None:
            zero = torch.zeros_like(
                inp_cnt if inp_cnt is not None
                else torch.zeros([2] + list(inp_res)))
            for k in ("inp_custom_cnt", "inp_custom_scaled_cnt",
                      "inp_custom_down_cnt", "inp_custom_down_scaled_cnt",
                      "gt_custom_cnt"):
                out[k] = zero.clone()

        if fields is not None:
            return {k: v for k, v in out.items()
                    if k in fields or k in ("gt_img", "gt_inp_size_img",
                                            "frame")}
        return out

    def _unsupervised_pair(self, norm_ev):
        """Down-scaled self-supervision pair (parity:
        ESR:dataloader/h5dataset.py:538-550)."""
        down_res = self.inp_down_sensor_resolution
        xs, ys, ts, ps = norm_ev
        down_ev = torch.stack([(xs * down_res[1]).long().float(),
                               (ys * down_res[0]).long().float(), ts, ps])
        down_norm = E.normalize_events(down_ev, down_res)
        s2 = self.scale ** 2
        inp_down_cnt = torch.div(
            E.scaled_count_encoding(down_norm, down_res, "cnt"), s2,
            rounding_mode="floor")
        inp_down_scaled_cnt = torch.div(
            E.scaled_count_encoding(down_norm, self.inp_sensor_resolution, "cnt"),
            s2, rounding_mode="floor")
        return inp_down_cnt, inp_down_scaled_cnt

    def _hot_mask(self, events, resolution):
        """Running-average hot-pixel mask (parity:
        ESR:dataloader/h5dataset.py:621-641, ESR:dataloader/encodings.py:348-363).
        Pixels whose cumulative event rate exceeds max_rate are zeroed, up
        to max_px pixels, once min_obvs windows have been observed."""
        update = torch.zeros(resolution)
        xi = events[0].long().clamp(0, resolution[1] - 1)
        yi = events[1].long().clamp(0, resolution[0] - 1)
        update[yi, xi] = events[3].abs()
        self.hot_events += update
        self.hot_idx += 1
        event_rate = self.hot_events / self.hot_idx
        mask = torch.ones(resolution)
        if self.hot_idx > self.hot_filter.get("min_obvs", 5):
            rate = event_rate.clone()
            for _ in range(self.hot_filter.get("max_px", 100)):
                idx = torch.argmax(rate)
                y, x = divmod(int(idx), resolution[1])
                if rate[y, x] > self.hot_filter.get("max_rate", 0.8):
                    rate[y, x] = 0
                    mask[y, x] = 0
                else:
                    break
        return mask

    def _get_gt_frame(self, idx0, idx1):
        """Frame nearest (by binary search) to the window's mid event
        (parity: ESR:dataloader/h5dataset.py:477-487)."""
        ref = (idx0 + idx1) // 2
        t = float(self.store.ts(self.inp_prefix)[ref])
        i = int(np.searchsorted(self.store.image_ts(), t, side="left"))
        i = min(max(i, 0), self.store.num_images - 1)
        return self.store.image(i)
