"""Checkpoint save/resume.

The on-disk structure is the reference's public checkpoint API
(ESR:train_ours_cnt_seq.py:642-658): a single torch.save dict

    {model: {name, states}, lr_scheduler: {name, states},
     optimizer: {name, states}, config, trainer: {training_mode,
     epoch|iteration, monitor_best}}

so inference can rebuild the model from the config stored inside the
checkpoint (ESR:infer_ours_cnt.py:118-132).  Resume verifies component
names before loading states (ESR:myutils/utils.py:140-177).
"""

from __future__ import annotations

import torch

__all__ = ["save_checkpoint", "Resumer", "load_model_from_checkpoint"]


def save_checkpoint(path, config: dict, model, optimizer=None,
                    lr_scheduler=None, trainer_state: dict | None = None):
    model_sd = model.module.state_dict() if hasattr(model, "module") \
        else model.state_dict()
    state = {
        "model": {"name": config["model"]["name"], "states": model_sd},
        "lr_scheduler": {"name": config["lr_scheduler"]["name"],
                         "states": lr_scheduler.state_dict() if lr_scheduler else {}},
        "optimizer": {"name": config["optimizer"]["name"],
                      "states": optimizer.state_dict() if optimizer else {}},
        "config": config,
    }
    if trainer_state is not None:
        state["trainer"] = trainer_state
    torch.save(state, path)
    return path


class Resumer:
    """Name-checked component state loading (parity:
    ESR:myutils/utils.py:140-177)."""

    def __init__(self, ckpt_path, logger=None, config: dict | None = None):
        self.ckpt = torch.load(ckpt_path, map_location="cpu", weights_only=False)
        self.logger = logger
        self.config = config or {}

    def _check(self, section):
        name_ckpt = self.ckpt[section]["name"]
        name_cfg = self.config.get(section, {}).get("name")
        if name_cfg is not None and name_cfg != name_ckpt:
            raise ValueError(
                f"checkpoint {section} '{name_ckpt}' != config '{name_cfg}'")

    def resume_trainer(self, section="trainer"):
        return self.ckpt.get(section, {})

    def resume_model(self, model, section="model"):
        self._check(section)
        target = model.module if hasattr(model, "module") else model
        target.load_state_dict(self.ckpt[section]["states"])

    def resume_optimizer(self, optimizer, section="optimizer"):
        self._check(section)
        if self.ckpt[section]["states"]:
            optimizer.load_state_dict(self.ckpt[section]["states"])

    def resume_lr_scheduler(self, sched, section="lr_scheduler"):
        self._check(section)
        if self.ckpt[section]["states"]:
            sched.load_state_dict(self.ckpt[section]["states"])


def load_model_from_checkpoint(ckpt_path, device="cpu", seqn: int | None = None):
    """Rebuild a model from the config stored in a checkpoint
    (parity: ESR:infer_ours_cnt.py:118-132)."""
    from ..models import build_model
    ckpt = torch.load(ckpt_path, map_location="cpu", weights_only=False)
    config = ckpt["config"]
    if seqn is not None and "SEQN" in config:
        assert config["SEQN"] == seqn, \
            f"checkpoint seqn {config['SEQN']} != requested {seqn}"
    model = build_model(ckpt["model"]["name"], **config["model"]["args"])
    model.load_state_dict(ckpt["model"]["states"])
    model.to(device).eval()
    return model, config
