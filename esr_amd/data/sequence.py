"""Sequence dataset: length-L runs of consecutive windows for BPTT, with an
optional Markov pause model (parity: ESR:dataloader/h5dataset.py:729-791).

Seeding divergence from the reference: the reference draws per-item seeds
from the worker-process RNG (ESR:dataloader/h5dataset.py:758), so the
SAME sampled index yields different augmentations depending on
num_workers and worker scheduling.  Here the seed is a stable hash of
(base seed, epoch, index): items are bit-identical for any worker count
(tests/test_data.py::test_loader_worker_count_invariance) while
augmentation still varies across epochs (`set_epoch`, called by the
Trainer alongside the sampler's)."""

from __future__ import annotations

import multiprocessing as mp
import random

from torch.utils.data import Dataset

from .dataset import EventSRDataset

__all__ = ["SequenceDataset"]


def _stable_seed(base: int, epoch: int, index: int) -> int:
    x = (base * 0x9E3779B1 + epoch * 0x85EBCA6B + index * 0xC2B2AE35) \
        & 0xFFFFFFFF
    x ^= x >> 16
    x = (x * 0x7FEB352D) & 0xFFFFFFFF
    x ^= x >> 15
    x = (x * 0x846CA68B) & 0xFFFFFFFF
    return x ^ (x >> 16)


class SequenceDataset(Dataset):
    def __init__(self, path, config: dict):
        super().__init__()
        self.config = config
        seq = config["sequence"]
        self.L = seq["sequence_length"]
        self.step_size = seq.get("step_size") or self.L
        self.pause_cfg = seq.get("pause", {"enabled": False})
        self.seed_base = int(config.get("seed", 123))
        # shared value: fork-inherited, so set_epoch() reaches live
        # persistent workers too (a plain attribute would be frozen in
        # the worker copies of the dataset)
        self._epoch = mp.Value("i", 0)

        assert self.L > 0 and self.step_size > 0
        self.dataset = EventSRDataset(path, config)
        if self.L >= self.dataset.length:
            self.length = 1
            self.L = self.dataset.length
        else:
            self.length = (self.dataset.length - self.L) // self.step_size + 1

        self.gt_sensor_resolution = self.dataset.gt_sensor_resolution
        self.inp_sensor_resolution = self.dataset.inp_sensor_resolution

    def __len__(self):
        return self.length

    def set_epoch(self, epoch: int):
        """Vary the per-index augmentation seed across epochs (the
        sampler-set_epoch analog for item randomness)."""
        with self._epoch.get_lock():
            self._epoch.value = int(epoch)

    def __getitem__(self, i):
        assert 0 <= i < self.length
        seed = _stable_seed(self.seed_base, self._epoch.value, i)
        rng = random.Random(seed ^ 0x5DEECE66D)   # pause chain, same seed
        j = i * self.step_size
        sequence = [self.dataset.__getitem__(j, seed=seed)]
        k = 0
        paused = False
        enabled = self.pause_cfg.get("enabled", False)
        for _ in range(self.L - 1):
            if enabled:
                u = rng.random()
                prob = (self.pause_cfg["proba_pause_when_paused"] if paused
                        else self.pause_cfg["proba_pause_when_running"])
                paused = u < prob
            if paused:
                sequence.append(self.dataset.__getitem__(j + k, pause=True, seed=seed))
            else:
                k += 1
                sequence.append(self.dataset.__getitem__(j + k, seed=seed))
        return sequence
