"""Cross-rank scalar reductions for logging (parity:
ESR:myutils/utils.py:42-82).  The reference issues a barrier before every
all-reduce; that is dropped — a scalar all-reduce is itself a sync point,
and on xGMI the extra barrier only adds a second latency-bound collective."""

from __future__ import annotations

import torch
import torch.distributed as dist

from .ddp import is_distributed


@torch.no_grad()
def reduce_tensor(t: torch.Tensor, average: bool = True) -> torch.Tensor:
    if not is_distributed() or dist.get_world_size() < 2:
        return t
    t = t.clone() if t.is_floating_point() else t
    dist.all_reduce(t)
    if average:
        t = t / dist.get_world_size()
    return t


@torch.no_grad()
def reduce_dict(d: dict, average: bool = True) -> dict:
    if not is_distributed() or dist.get_world_size() < 2:
        return d
    names = sorted(d.keys())
    values = torch.stack([torch.as_tensor(d[k]).float() for k in names])
    dist.all_reduce(values)
    if average:
        values /= dist.get_world_size()
    return {k: v for k, v in zip(names, values)}
