"""Sequence dataset: length-L runs of consecutive windows for BPTT, with an
optional Markov pause model (parity: ESR:dataloader/h5dataset.py:729-791)."""

from __future__ import annotations

import random

from torch.utils.data import Dataset

from .dataset import EventSRDataset

__all__ = ["SequenceDataset"]


class SequenceDataset(Dataset):
    def __init__(self, path, config: dict):
        super().__init__()
        self.config = config
        seq = config["sequence"]
        self.L = seq["sequence_length"]
        self.step_size = seq.get("step_size") or self.L
        self.pause_cfg = seq.get("pause", {"enabled": False})

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

    def __getitem__(self, i):
        assert 0 <= i < self.length
        seed = random.randint(0, 2 ** 32)
        j = i * self.step_size
        sequence = [self.dataset.__getitem__(j, seed=seed)]
        k = 0
        paused = False
        enabled = self.pause_cfg.get("enabled", False)
        for _ in range(self.L - 1):
            if enabled:
                u = random.random()
                prob = (self.pause_cfg["proba_pause_when_paused"] if paused
                        else self.pause_cfg["proba_pause_when_running"])
                paused = u < prob
            if paused:
                sequence.append(self.dataset.__getitem__(j + k, pause=True, seed=seed))
            else:
                k += 1
                sequence.append(self.dataset.__getitem__(j + k, seed=seed))
        return sequence
