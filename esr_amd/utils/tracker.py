"""Running-average metric tracker (parity: ESR:myutils/utils.py:85-106,
without the pandas dependency)."""

from __future__ import annotations


class MetricTracker:
    def __init__(self, keys, writer=None):
        self.writer = writer
        self._total = {k: 0.0 for k in keys}
        self._count = {k: 0 for k in keys}

    def reset(self):
        for k in self._total:
            self._total[k] = 0.0
            self._count[k] = 0

    def update(self, key, value, n=1):
        if self.writer is not None:
            self.writer.add_scalar(key, value)
        self._total[key] += value * n
        self._count[key] += n

    def avg(self, key):
        c = self._count[key]
        return self._total[key] / c if c else 0.0

    def result(self):
        return {k: self.avg(k) for k in self._total}
