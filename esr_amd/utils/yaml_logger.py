"""YAML results logger (parity: ESR:myutils/utils.py:180-192)."""

from __future__ import annotations

import yaml


class YamlLogger:
    def __init__(self, path: str):
        self.path = path
        self._data = {}

    def log_dict(self, d: dict, name: str):
        self._data[name] = d
        self._flush()

    def log_info(self, info: str):
        self._data.setdefault("info", []).append(info)
        self._flush()

    def _flush(self):
        with open(self.path, "w") as f:
            yaml.safe_dump(self._data, f, default_flow_style=False)
