"""Wall-clock and device-event timers (parity: ESR:myutils/timers.py:29-77).

``CudaTimer`` uses hipEvent pairs through ``torch.cuda.Event`` — the
reference left its CUDA-event timer commented out and timed with host
wall-clock only; on MI355X the event timer is the accurate per-op tool.
Aggregated stats are printed by ``print_timing_info`` (the reference does
this via atexit).
"""

from __future__ import annotations

import atexit
import time
from collections import defaultdict

import torch

cuda_timers = defaultdict(list)
timers = defaultdict(list)


class CudaTimer:
    def __init__(self, timer_name=""):
        self.timer_name = timer_name
        self.start_ev = torch.cuda.Event(enable_timing=True)
        self.end_ev = torch.cuda.Event(enable_timing=True)

    def __enter__(self):
        self.start_ev.record()
        return self

    def __exit__(self, *args):
        self.end_ev.record()
        self.end_ev.synchronize()
        cuda_timers[self.timer_name].append(
            self.start_ev.elapsed_time(self.end_ev))


class Timer:
    def __init__(self, timer_name="", logger=None):
        self.timer_name = timer_name
        self.logger = logger

    def __enter__(self):
        self.start = time.time()
        return self

    def __exit__(self, *args):
        self.interval = (time.time() - self.start) * 1000.0  # ms
        timers[self.timer_name].append(self.interval)
        if self.logger is not None:
            self.logger.info(f"{self.timer_name}: {self.interval:.1f} ms")


def print_timing_info():
    for name, vals in list(timers.items()) + list(cuda_timers.items()):
        if vals:
            print(f"[timer] {name}: mean {sum(vals)/len(vals):.3f} ms over {len(vals)} calls")


atexit.register(print_timing_info)
