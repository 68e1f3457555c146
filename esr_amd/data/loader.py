"""Data loaders with sequence collate and sharded (distributed) sampling.

Parity: ESR:dataloader/h5dataloader.py:20-347.  The sequence collate turns a
batch of length-L window sequences into seql-seqn+1 sliding BPTT windows,
each a dict of [B, seqn, ...] tensors (ESR:dataloader/h5dataloader.py:210-246).
"""

from __future__ import annotations

from collections import defaultdict

import torch
from torch.utils.data import ConcatDataset, DataLoader
from torch.utils.data.distributed import DistributedSampler

from ..parallel import is_distributed
from .dataset import EventSRDataset
from .sequence import SequenceDataset

__all__ = ["read_datalist", "concatenate_datasets", "make_event_loader",
           "SequenceDataLoader", "InferenceSequenceDataLoader",
           "sequence_collate", "shared_sequence_collate"]

_PACK_KEYS = {"inp_events", "inp_normalized_events", "inp_scaled_events",
              "inp_pol_mask", "gt_events", "gt_normalized_events"}


def read_datalist(path: str) -> list[str]:
    """A datalist txt is one sequence path per line (parity:
    ESR:dataloader/h5dataloader.py:28)."""
    with open(path) as f:
        return [ln.strip() for ln in f if ln.strip() and not ln.startswith("#")]


def concatenate_datasets(datalist_path, dataset_type, dataset_config):
    paths = read_datalist(datalist_path)
    return ConcatDataset([dataset_type(p, dataset_config) for p in paths])


def pack_tensor(seq):
    """Zero-pad a list of [N_i, C] tensors to [B, N_max, C] (parity:
    ESR:dataloader/h5dataloader.py:248-263)."""
    maxlen = max(item.size(0) for item in seq)
    out = torch.zeros(len(seq), maxlen, seq[0].size(1))
    for i, item in enumerate(seq):
        out[i, : item.size(0)] = item
    return out


def _stack_items(items: list[dict]) -> dict:
    body = {}
    for key in items[0].keys():
        vals = [it[key] for it in items]
        body[key] = pack_tensor(vals) if key in _PACK_KEYS else torch.stack(vals)
    return body


def flat_collate(batch: list[dict]) -> dict:
    return _stack_items(batch)


def _concat_dict(dicts: list[dict]) -> dict:
    out = defaultdict(list)
    for key in dicts[0].keys():
        for d in dicts:
            out[key].append(d[key])
    return {k: torch.stack(v, dim=1) for k, v in out.items()}


class sequence_collate:
    """Collate for SequenceDataset batches; picklable for worker processes."""

    def __init__(self, seqn: int):
        self.seqn = seqn

    def __call__(self, batch: list[list[dict]]) -> list[dict]:
        L = len(batch[0])
        per_step = [_stack_items([entry[i] for entry in batch]) for i in range(L)]
        assert L >= self.seqn
        return [_concat_dict(per_step[i:i + self.seqn])
                for i in range(L - self.seqn + 1)]


def _make_loader(dataset, cfg, collate_fn):
    use_ddp = cfg.get("use_ddp", False) and is_distributed()
    common = dict(batch_size=cfg["batch_size"], num_workers=cfg["num_workers"],
                  pin_memory=cfg["pin_memory"], drop_last=cfg["drop_last"],
                  collate_fn=collate_fn,
                  persistent_workers=cfg["num_workers"] > 0)
    if use_ddp:
        sampler = DistributedSampler(dataset, shuffle=cfg["shuffle"])
        return DataLoader(dataset, sampler=sampler, **common), sampler
    return DataLoader(dataset, shuffle=cfg["shuffle"], **common), None


def make_event_loader(dataloader_config):
    """Flat (non-sequence) loader over a datalist
    (parity: ESR:dataloader/h5dataloader.py:38-67)."""
    ds = concatenate_datasets(dataloader_config["path_to_datalist_txt"],
                              EventSRDataset, dataloader_config["dataset"])
    loader, sampler = _make_loader(ds, dataloader_config, flat_collate)
    loader.gt_sensor_resolution = ds.datasets[0].gt_sensor_resolution
    loader.inp_sensor_resolution = ds.datasets[0].inp_sensor_resolution
    loader.scale = dataloader_config["dataset"]["scale"]
    loader.dist_sampler = sampler
    return loader


def SequenceDataLoader(dataloader_config):
    """Sequence loader over a datalist (parity:
    ESR:dataloader/h5dataloader.py:180-208).  Returns a DataLoader with
    .seqn/.dist_sampler/.inp_sensor_resolution/.gt_sensor_resolution set.
    collate='shared' yields shared-encoder sequence batches instead of
    materialized windows."""
    ds = concatenate_datasets(dataloader_config["path_to_datalist_txt"],
                              SequenceDataset, dataloader_config["dataset"])
    seqn = dataloader_config["dataset"]["sequence"]["seqn"]
    collate_cls = shared_sequence_collate \
        if dataloader_config.get("collate") == "shared" else sequence_collate
    loader, sampler = _make_loader(ds, dataloader_config, collate_cls(seqn))
    loader.seqn = seqn
    loader.dist_sampler = sampler
    loader.gt_sensor_resolution = ds.datasets[0].gt_sensor_resolution
    loader.inp_sensor_resolution = ds.datasets[0].inp_sensor_resolution
    loader.scale = dataloader_config["dataset"]["scale"]

    def set_epoch(epoch: int):
        """Advance the per-index augmentation seeds (and the sharded
        sampler's shuffle) — item randomness is epoch-keyed, not
        worker-keyed, so runs reproduce for any num_workers.
        NOTE: with persistent_workers the new epoch takes effect at the
        next iterator creation (workers re-pickle the dataset)."""
        for d in ds.datasets:
            d.set_epoch(epoch)
        if sampler is not None:
            sampler.set_epoch(epoch)
    loader.set_epoch = set_epoch
    return loader


def InferenceSequenceDataLoader(data_path, dataloader_config):
    """Single-sequence inference loader (parity:
    ESR:dataloader/h5dataloader.py:271-312)."""
    ds = SequenceDataset(data_path, dataloader_config["dataset"])
    seqn = dataloader_config["dataset"]["sequence"]["seqn"]
    cfg = dict(dataloader_config)
    cfg["use_ddp"] = False
    loader, _ = _make_loader(ds, cfg, sequence_collate(seqn))
    loader.seqn = seqn
    loader.dist_sampler = None
    loader.gt_sensor_resolution = ds.gt_sensor_resolution
    loader.inp_sensor_resolution = ds.inp_sensor_resolution
    loader.scale = dataloader_config["dataset"]["scale"]
    return loader


class shared_sequence_collate:
    """Collate for shared-encoder BPTT (ESRNet.forward_sequence): returns
    one dict per batch with the L UNIQUE window frames plus the per-window
    mid-frame ground truths, instead of L-seqn+1 materialized windows."""

    def __init__(self, seqn: int):
        self.seqn = seqn
        self.mid_idx = (seqn - 1) // 2

    def __call__(self, batch: list[list[dict]]) -> dict:
        L = len(batch[0])
        per_step = [_stack_items([entry[i] for entry in batch])
                    for i in range(L)]
        assert L >= self.seqn
        n_windows = L - self.seqn + 1
        frames = torch.stack([s["inp_scaled_cnt"] for s in per_step], dim=1)
        gt_mids = torch.stack(
            [per_step[w + self.mid_idx]["gt_cnt"] for w in range(n_windows)],
            dim=1)
        return {"frames": frames,            # [B, L, 2, kH, kW]
                "gt_mids": gt_mids,          # [B, n_windows, 2, kH, kW]
                "seqn": self.seqn,
                "mid_step": per_step[self.mid_idx]}
