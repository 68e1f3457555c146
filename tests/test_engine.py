"""Trainer / checkpoint / inference engine tests (CPU)."""

import math
from pathlib import Path

import pytest
import torch
import yaml

from esr_amd.config import ConfigParser
from esr_amd.engine import build_training, load_model_from_checkpoint
from esr_amd.engine.inference import build_metrics, infer_sequence
from esr_amd.utils.logging import setup_logging


def _train_config(datalist, out_dir, iterations=2):
    ds = {
        "scale": 2, "ori_scale": "down4", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 1024, "sliding_window": 512,
        "data_augment": {"enabled": False, "augment": [], "augment_prob": []},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 4, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0.05,
                               "proba_pause_when_paused": 0.9}},
    }
    dl = {"use_ddp": False, "path_to_datalist_txt": str(datalist),
          "batch_size": 1, "shuffle": False, "num_workers": 0,
          "pin_memory": False, "drop_last": True, "dataset": ds}
    return {
        "experiment": "test",
        "SEQN": 3,
        "model": {"name": "ESRNet",
                  "args": {"inch": 2, "basech": 4, "num_frame": 3}},
        "optimizer": {"name": "Adam", "args": {"lr": 1e-3}},
        "lr_scheduler": {"name": "ExponentialLR", "args": {"gamma": 0.95}},
        "trainer": {
            "output_path": str(out_dir),
            "epoch_based_train": {"enabled": False},
            "iteration_based_train": {
                "enabled": True, "iterations": iterations, "save_period": 1,
                "train_log_step": 1, "valid_log_step": 1, "valid_step": 1,
                "lr_change_rate": 1000},
            "monitor": "min valid_loss", "early_stop": 10,
            "tensorboard": False,
            "vis": {"enabled": False},
        },
        "train_dataloader": dl,
        "valid_dataloader": dict(dl),
    }


@pytest.fixture()
def trained(tmp_path, synth_datalist):
    cfg = _train_config(synth_datalist, tmp_path / "out", iterations=2)
    parser = ConfigParser(cfg, run_id="t0")
    logger = setup_logging("test", None)
    trainer = build_training(parser, torch.device("cpu"), logger)
    trainer.train()
    return parser, trainer


def test_training_runs_and_checkpoints(trained):
    parser, trainer = trained
    ckpts = list(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))
    assert ckpts, "no checkpoint written"
    state = torch.load(ckpts[0], map_location="cpu", weights_only=False)
    # reference-compatible checkpoint structure
    # (ESR:train_ours_cnt_seq.py:642-658)
    assert set(state) >= {"model", "lr_scheduler", "optimizer", "config", "trainer"}
    assert state["model"]["name"] == "ESRNet"
    assert "states" in state["model"]
    assert state["trainer"]["training_mode"] == "iteration_based_train"
    assert math.isfinite(state["trainer"]["monitor_best"])


def test_resume(trained, synth_datalist, tmp_path):
    parser, trainer = trained
    ckpt = sorted(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))[-1]
    cfg = _train_config(synth_datalist, tmp_path / "out2", iterations=2)
    parser2 = ConfigParser(cfg, run_id="t1")
    logger = setup_logging("test2", None)
    trainer2 = build_training(parser2, torch.device("cpu"), logger,
                              resume=str(ckpt))
    assert trainer2.start_iteration >= 1
    # weights actually loaded
    p1 = next(trainer.model.parameters())
    p2 = next(trainer2.model.parameters())
    assert torch.allclose(p1, p2)


def test_inference_harness(trained, synth_datalist, tmp_path):
    parser, trainer = trained
    ckpt = sorted(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))[-1]
    model, cfg = load_model_from_checkpoint(ckpt, device="cpu", seqn=3)
    from esr_amd.data import read_datalist
    dl_cfg = {"batch_size": 1, "shuffle": False, "num_workers": 0,
              "pin_memory": False, "drop_last": False, "use_ddp": False,
              "dataset": _train_config(synth_datalist, tmp_path)["train_dataloader"]["dataset"]}
    out = tmp_path / "infer"
    metrics = build_metrics("cpu")
    res = infer_sequence(dl_cfg, read_datalist(synth_datalist)[0], model,
                         torch.device("cpu"), output_path=out,
                         metrics=metrics, save_images=True, max_batches=2)
    for k in ["esr_mse", "bicubic_mse", "esr_ssim", "esr_psnr", "esr_lpips",
              "time", "params"]:
        assert k in res and math.isfinite(res[k])
    assert (out / "results.yml").exists()
    pngs = list((out / "event_imgs" / "hr_esr_event_img").glob("*.png"))
    assert pngs, "no visualization written"


def test_monitor_early_stop_logic(trained):
    _, trainer = trained
    trainer.mnt_best = 0.0
    trainer.not_improved_count = 0
    trainer.early_stop = 1
    stop, best = trainer.eval_model_performance({"valid_loss": 1.0})
    assert not stop and not best
    stop, best = trainer.eval_model_performance({"valid_loss": 2.0})
    assert stop


def test_shared_collate_training(tmp_path, synth_datalist):
    """collate='shared' trains through forward_sequence with the same
    machinery (CPU, single process)."""
    cfg = _train_config(synth_datalist, tmp_path / "out_shared", iterations=2)
    cfg["train_dataloader"]["collate"] = "shared"
    cfg["valid_dataloader"]["collate"] = "shared"
    parser = ConfigParser(cfg, run_id="sh0")
    logger = setup_logging("test-shared", None)
    trainer = build_training(parser, torch.device("cpu"), logger)
    trainer.train()
    assert math.isfinite(trainer.train_metrics.avg("train_loss"))
    assert trainer.train_metrics.avg("train_loss") > 0


def test_epoch_based_training(tmp_path, synth_datalist):
    cfg = _train_config(synth_datalist, tmp_path / "out_epoch")
    cfg["trainer"]["iteration_based_train"] = {"enabled": False}
    cfg["trainer"]["epoch_based_train"] = {
        "enabled": True, "epochs": 2, "save_period": 1,
        "train_log_step": 1, "valid_log_step": 1, "valid_step": 1}
    parser = ConfigParser(cfg, run_id="ep0")
    logger = setup_logging("test-epoch", None)
    trainer = build_training(parser, torch.device("cpu"), logger)
    trainer.train()
    ckpts = list(Path(parser.save_dir).glob("checkpoint-epoch*.pth"))
    assert ckpts, "no epoch checkpoint written"
    state = torch.load(ckpts[-1], map_location="cpu", weights_only=False)
    assert state["trainer"]["training_mode"] == "epoch_based_train"


def test_resumer_rejects_wrong_component_name(trained, tmp_path):
    from esr_amd.engine.checkpoint import Resumer
    parser, trainer = trained
    ckpt = sorted(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))[-1]
    bad_cfg = {"model": {"name": "SomethingElse"}}
    r = Resumer(str(ckpt), config=bad_cfg)
    with pytest.raises(ValueError):
        r.resume_model(trainer.model)
