"""Distributed failure detection.

The reference has none: a dead rank hangs the job at the next barrier
forever (SURVEY §5 — ESR:train_ours_cnt_seq.py:339).  Here every rank
heartbeats into the torch.distributed TCPStore; a monitor thread flags
ranks whose heartbeat goes stale and can abort the local process group so
the launcher (torchrun) can tear the job down and it can be relaunched
from the last checkpoint (esr_amd.engine.checkpoint).
"""

from __future__ import annotations

import os
import threading
import time

import torch.distributed as dist

__all__ = ["Watchdog"]


class Watchdog:
    """Store-based heartbeat failure detector.

    Usage (per rank, after init_process_group):
        wd = Watchdog(timeout=60.0).start()
        ...training loop... wd.beat() every iteration ...
        wd.stop()

    If any peer's heartbeat is older than `timeout` seconds, `failed_ranks`
    becomes non-empty and `on_failure` (default: log + hard-abort via
    os._exit so the launcher notices) is invoked on the monitor thread.
    """

    def __init__(self, timeout: float = 60.0, interval: float = 5.0,
                 store=None, on_failure=None):
        if not (dist.is_available() and dist.is_initialized()):
            raise RuntimeError("Watchdog requires an initialized process group")
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.timeout = timeout
        self.interval = interval
        self.store = store or dist.distributed_c10d._get_default_store()
        self.on_failure = on_failure or self._default_on_failure
        self.failed_ranks: list[int] = []
        self._stop = threading.Event()
        self._thread = None

    def _key(self, rank):
        return f"esr_watchdog/hb/{rank}"

    def beat(self):
        self.store.set(self._key(self.rank), str(time.time()))

    def check(self) -> list[int]:
        now = time.time()
        stale = []
        for r in range(self.world):
            if r == self.rank:
                continue
            try:
                t = float(self.store.get(self._key(r)))
            except Exception:
                continue  # peer has not heartbeat yet
            if now - t > self.timeout:
                stale.append(r)
        return stale

    def _default_on_failure(self, ranks):
        import logging
        logging.getLogger("esr.watchdog").error(
            f"rank {self.rank}: peers {ranks} missed heartbeats for "
            f">{self.timeout}s; aborting so the launcher can restart "
            f"from the last checkpoint")
        os._exit(134)

    def _loop(self):
        while not self._stop.wait(self.interval):
            self.beat()
            stale = self.check()
            if stale:
                self.failed_ranks = stale
                self.on_failure(stale)
                return

    def start(self):
        self.beat()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2 * self.interval)
