"""Small tensor utilities (parity: ESR:myutils/utils.py:13-39, :108-112)."""

from __future__ import annotations

import torch


def normalize_nonzero(x: torch.Tensor) -> torch.Tensor:
    """Zero-mean/unit-std normalize only the NONZERO entries of an event
    tensor (parity: ESR:myutils/utils.py:13-31)."""
    nonzero = x != 0
    n = nonzero.sum()
    if n > 0:
        mean = x.sum() / n
        std = torch.sqrt((x ** 2).sum() / n - mean ** 2)
        x = nonzero.float() * (x - mean) / (std + 1e-12)
    return x


def inf_loop(data_loader):
    """Endless dataloader wrapper (parity: ESR:myutils/utils.py:108-112)."""
    while True:
        yield from data_loader
