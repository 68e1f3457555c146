"""Streaming (serving) inference.

The reference only has offline file evaluation (ESR:infer_ours_cnt.py).
This is the production path: a persistent pipeline that consumes raw event
windows, splats them onto the HR grid on-device with the native kernels,
runs the recurrent model with state carried across windows (the recurrence
IS the serving state), and returns HR count maps.  On GPU the forward is
hipGraph-captured after the first window, so steady-state serving replays
one graph per window.
"""

from __future__ import annotations

from collections import deque

import torch

from ..ops import events as E
from ..ops.native import get_ext

__all__ = ["StreamingESR"]


class StreamingESR:
    """Stateful sliding-window SR server.

    Feed windows of LR events ([4, n] float tensors: x, y, t, p in LR pixel
    coordinates); once `seqn` windows have arrived every further window
    yields one HR count map [2, kH, kW].
    """

    def __init__(self, model, lr_resolution, scale: int = 2, seqn: int = 3,
                 device="cuda:0", use_graphs: bool = True,
                 amp_dtype=torch.bfloat16):
        self.model = model.to(device).eval()
        self.device = torch.device(device)
        self.lr_res = tuple(lr_resolution)
        self.hr_res = (self.lr_res[0] * scale, self.lr_res[1] * scale)
        self.scale = scale
        self.seqn = seqn
        self.amp_dtype = amp_dtype
        self.use_graphs = use_graphs and self.device.type == "cuda"
        self._frames: deque[torch.Tensor] = deque(maxlen=seqn)
        self._graph = None
        self._static_in = None
        self._static_out = None
        if hasattr(self.model, "reset_states"):
            self.model.reset_states()

    def reset(self):
        self._frames.clear()
        if hasattr(self.model, "reset_states"):
            self.model.reset_states()

    # ------------------------------------------------------------------
    def _splat(self, events: torch.Tensor) -> torch.Tensor:
        """LR events [4, n] -> HR count map [2, kH, kW] on device."""
        ev = events.to(self.device, non_blocking=True).float()
        xs = torch.div(ev[0], 1, rounding_mode="floor") * self.scale
        ys = torch.div(ev[1], 1, rounding_mode="floor") * self.scale
        ext = get_ext()
        if ext is not None and self.device.type == "cuda":
            packed = torch.stack([xs, ys, ev[2], ev[3]], dim=1)[None] \
                .contiguous()
            return ext.splat_count(packed, *self.hr_res)[0]
        return E.events_to_channels(xs, ys, ev[3], self.hr_res)

    @torch.no_grad()
    def _forward(self, inp: torch.Tensor) -> torch.Tensor:
        if self.amp_dtype is not None and self.device.type == "cuda":
            with torch.autocast("cuda", dtype=self.amp_dtype,
                                cache_enabled=False):
                return self.model(inp).float()
        return self.model(inp)

    def _state_holder(self):
        inner = self.model.module if hasattr(self.model, "module") else self.model
        return getattr(inner, "time_propagate", None)

    def _graph_body(self):
        # recurrence across replays: the state lives in one static buffer
        # that the captured region reads AND writes back in place
        holder = self._state_holder()
        holder.state = self._static_state
        out = self._forward(self._static_in)
        self._static_state.copy_(holder.state)
        holder.state = self._static_state
        return out

    @torch.no_grad()
    def _forward_graphed(self, inp: torch.Tensor) -> torch.Tensor:
        holder = self._state_holder()
        if holder is None:
            raise RuntimeError("model has no recurrent state holder")
        if self._graph is None:
            self._static_in = inp.clone()
            # run once eagerly to materialize the state shape
            out = self._forward(self._static_in)
            self._static_state = holder.state.detach().clone()
            # warmup and capture both EXECUTE the body, each advancing the
            # in-place state buffer — snapshot and restore the live state
            state_snapshot = self._static_state.clone()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self._graph_body()
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._static_out = self._graph_body()
            torch.cuda.synchronize()
            self._static_state.copy_(state_snapshot)
            holder.state = self._static_state
            self._graph = g
            return out
        self._static_in.copy_(inp)
        self._graph.replay()
        return self._static_out

    @torch.no_grad()
    def push(self, events: torch.Tensor) -> torch.Tensor | None:
        """Push one LR event window; returns the HR count map once warm."""
        self._frames.append(self._splat(events))
        if len(self._frames) < self.seqn:
            return None
        inp = torch.stack(tuple(self._frames))[None]     # [1, seqn, 2, kH, kW]
        if self.use_graphs:
            try:
                # clone: the graph writes into one static output buffer,
                # the caller gets an independent tensor
                return self._forward_graphed(inp)[0].clone()
            except Exception:
                self.use_graphs = False
        return self._forward(inp)[0]

    @torch.no_grad()
    def push_events(self, events: torch.Tensor,
                    capacity: int | None = None,
                    mode: str = "linear") -> torch.Tensor | None:
        """Push one LR event window and get the prediction back AS AN HR
        EVENT STREAM [n, 4] (x, y, t, p sorted by t) — the serving-side
        count->event conversion (reference cnt2event,
        ESR:dataloader/cython_cnt2event/cnt2event.pyx:18-116).  On GPU this
        runs the device-only redistribution pipeline (redistribute.hip);
        pass `capacity` for a fixed-shape, sync-free conversion."""
        cnt = self.push(events)
        if cnt is None:
            return None
        ev = E.redistribute_count(cnt.float().round().clamp(min=0)[None],
                                  mode=mode, capacity=capacity)[0]
        n = int((ev[:, 3] != 0).sum()) if capacity is None else capacity
        return ev[:n]
