"""YAML configuration system (parity: ESR:config/parser.py:14-147).

Differences from the reference: components are instantiated through explicit
registries (models, optimizers, schedulers) instead of ``eval()`` of config
strings, and the parser is usable without creating run directories (tests).
"""

from __future__ import annotations

import argparse
import time
from pathlib import Path

import yaml

OPTIMIZERS = {
    "Adam": "Adam", "AdamW": "AdamW", "SGD": "SGD", "RMSprop": "RMSprop",
}
SCHEDULERS = {
    "ExponentialLR": "ExponentialLR", "StepLR": "StepLR",
    "CosineAnnealingLR": "CosineAnnealingLR", "MultiStepLR": "MultiStepLR",
}


def build_optimizer(name: str, params, **kwargs):
    import torch.optim as optim
    if name not in OPTIMIZERS:
        raise KeyError(f"unknown optimizer '{name}'; known: {sorted(OPTIMIZERS)}")
    return getattr(optim, OPTIMIZERS[name])(params, **kwargs)


def build_lr_scheduler(name: str, optimizer, **kwargs):
    import torch.optim.lr_scheduler as sched
    if name not in SCHEDULERS:
        raise KeyError(f"unknown scheduler '{name}'; known: {sorted(SCHEDULERS)}")
    return getattr(sched, SCHEDULERS[name])(optimizer, **kwargs)


def set_by_path(tree: dict, keys: str, value):
    """Set a nested key addressed as 'a;b;c' (parity: ESR:config/parser.py:103-112)."""
    parts = keys.split(";")
    node = tree
    for k in parts[:-1]:
        node = node.setdefault(k, {})
    node[parts[-1]] = value


def get_by_path(tree: dict, keys: str, default=None):
    node = tree
    for k in keys.split(";"):
        if not isinstance(node, dict) or k not in node:
            return default
        node = node[k]
    return node


class ConfigParser:
    """Loads a YAML config, optionally creates run dirs + a config snapshot,
    and applies CLI flag overrides addressed by 'a;b;c' key paths."""

    def __init__(self, config: dict, run_id: str | None = None,
                 output_path: str | None = None, make_dirs: bool = True,
                 args: argparse.Namespace | None = None):
        self.config = config
        self.args = args
        run_id = run_id or time.strftime("%m%d_%H%M%S")
        self.run_id = run_id

        out = output_path or get_by_path(config, "trainer;output_path")
        if make_dirs and out:
            base = Path(out) / run_id
            self.save_dir = base / "models"
            self.log_dir = base / "log"
            self.save_dir.mkdir(parents=True, exist_ok=True)
            self.log_dir.mkdir(parents=True, exist_ok=True)
            with open(base / "config.yml", "w") as f:
                yaml.safe_dump(self.config, f, default_flow_style=False)
        else:
            self.save_dir = None
            self.log_dir = None

    # dict-like access (parity: ESR:config/parser.py:82-84)
    def __getitem__(self, key):
        return self.config[key]

    def __contains__(self, key):
        return key in self.config

    def get(self, key, default=None):
        return self.config.get(key, default)

    @classmethod
    def from_file(cls, path: str, run_id=None, overrides=None,
                  make_dirs: bool = True, args=None) -> "ConfigParser":
        with open(path) as f:
            config = yaml.safe_load(f)
        for keys, value in (overrides or {}).items():
            set_by_path(config, keys, value)
        return cls(config, run_id=run_id, make_dirs=make_dirs, args=args)

    @classmethod
    def from_args(cls, parser: argparse.ArgumentParser, options=()):
        """Build from CLI args; `options` are (flags, type, target) triples
        whose values override nested config keys (ESR:config/parser.py:46-61)."""
        for opt in options:
            parser.add_argument(*opt.flags, default=None, type=opt.type)
        args = parser.parse_args()
        overrides = {}
        for opt in options:
            name = opt.flags[-1].lstrip("-").replace("-", "_")
            val = getattr(args, name, None)
            if val is not None:
                overrides[opt.target] = val
        cfg = cls.from_file(args.config, run_id=getattr(args, "runid", None),
                            overrides=overrides, args=args)
        return cfg

    def get_logger(self, name: str):
        from ..utils.logging import setup_logging
        return setup_logging(name, self.log_dir)
