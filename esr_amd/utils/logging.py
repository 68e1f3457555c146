"""Logging & scalar/metric recording.

Replaces the reference's JSON-dictConfig logging + TensorBoard
(ESR:logger/logger.py:8-24, ESR:logger/visualization.py:5-73) with a
self-contained console+file logger and a JSONL scalar writer (tensorboard is
not installed in this environment; if it is importable it is used as an
additional sink).
"""

from __future__ import annotations

import json
import logging
import time
from pathlib import Path

_LOGGERS: dict[str, logging.Logger] = {}


def setup_logging(name: str = "esr", log_dir=None,
                  level=logging.INFO) -> logging.Logger:
    key = f"{name}:{log_dir}"
    if key in _LOGGERS:
        return _LOGGERS[key]
    logger = logging.getLogger(name)
    logger.setLevel(level)
    logger.propagate = False
    if not logger.handlers:
        fmt = logging.Formatter("%(asctime)s %(levelname)s %(name)s: %(message)s")
        sh = logging.StreamHandler()
        sh.setFormatter(fmt)
        logger.addHandler(sh)
        if log_dir is not None:
            Path(log_dir).mkdir(parents=True, exist_ok=True)
            fh = logging.FileHandler(Path(log_dir) / "info.txt")
            fh.setFormatter(fmt)
            logger.addHandler(fh)
    _LOGGERS[key] = logger
    return logger


class MetricWriter:
    """Scalar writer: JSONL file (+TensorBoard when available).

    API mirrors the slice of TensorboardWriter the trainer uses
    (ESR:logger/visualization.py:5-73): set_step(), add_scalar(),
    add_image() (image adds are JSONL-skipped), steps/sec is derived from
    set_step timing.
    """

    def __init__(self, log_dir=None, enabled=True):
        self.enabled = enabled and log_dir is not None
        self.step = 0
        self.mode = "train"
        self._f = None
        self._tb = None
        self._t_prev = None
        if self.enabled:
            Path(log_dir).mkdir(parents=True, exist_ok=True)
            self._f = open(Path(log_dir) / "scalars.jsonl", "a")
            try:
                from torch.utils.tensorboard import SummaryWriter  # optional
                self._tb = SummaryWriter(str(log_dir))
            except Exception:
                self._tb = None

    def set_step(self, step, mode="train"):
        self.mode = mode
        self.step = step
        now = time.time()
        if mode == "train":
            if self._t_prev is not None and step > self._t_prev[0]:
                sps = (step - self._t_prev[0]) / (now - self._t_prev[1])
                self.add_scalar("steps_per_sec", sps)
            self._t_prev = (step, now)

    def add_scalar(self, tag, value, global_step=None):
        if not self.enabled:
            return
        step = self.step if global_step is None else global_step
        rec = {"t": time.time(), "mode": self.mode, "step": step,
               "tag": tag, "value": float(value)}
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()
        if self._tb is not None:
            self._tb.add_scalar(f"{self.mode}/{tag}", value, step)

    def add_image(self, tag, img, global_step=None, dataformats="HWC"):
        if self._tb is not None:
            self._tb.add_image(tag, img, global_step or self.step,
                               dataformats=dataformats)

    def close(self):
        if self._f:
            self._f.close()
        if self._tb:
            self._tb.close()
