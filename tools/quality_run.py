#!/usr/bin/env python3
"""Train briefly on ESIM-simulated data and evaluate ESR vs bicubic.

Produces the reference's evaluation artifact (mean metric table, YAML) on
synthetic data — evidence that the training loop learns and the model
beats the bicubic baseline on the reconstruction metrics.

  python tools/quality_run.py --out runs/quality --iterations 300
"""

import argparse
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402
import yaml  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="runs/quality")
    p.add_argument("--iterations", type=int, default=300)
    p.add_argument("--basech", type=int, default=8)
    p.add_argument("--resolution", type=int, default=128)
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available()
                   else "cpu")
    args = p.parse_args()

    import os
    os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

    from esr_amd.config import ConfigParser
    from esr_amd.data.synthetic import make_synthetic_dataset
    from esr_amd.engine import build_training
    from esr_amd.engine.inference import build_metrics, infer_sequence
    from esr_amd.utils.logging import setup_logging

    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    datalist = make_synthetic_dataset(out / "data", num_sequences=3,
                                      resolution=(args.resolution,) * 2,
                                      num_events=400_000, seed=3)

    ds = {
        "scale": 2, "ori_scale": "down2", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 2048, "sliding_window": 1024,
        "data_augment": {"enabled": True,
                         "augment": ["Horizontal", "Vertical", "Polarity"],
                         "augment_prob": [0.5, 0.5, 0.5]},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 5, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0.05,
                               "proba_pause_when_paused": 0.9}},
    }
    dl = {"use_ddp": False, "path_to_datalist_txt": datalist,
          "batch_size": 4, "shuffle": True, "num_workers": 2,
          "pin_memory": True, "drop_last": True, "dataset": ds}
    cfg = {
        "experiment": "quality-run", "SEQN": 3, "precision": "bf16",
        "model": {"name": "ESRNet",
                  "args": {"inch": 2, "basech": args.basech, "num_frame": 3,
                           "upsampler": "pixelshuffle"}},
        "optimizer": {"name": "Adam",
                      "args": {"lr": 1e-3, "weight_decay": 1e-4,
                               "amsgrad": True}},
        "lr_scheduler": {"name": "ExponentialLR", "args": {"gamma": 0.95}},
        "trainer": {"output_path": str(out / "train"),
                    "hip_graphs": torch.cuda.is_available(),
                    "epoch_based_train": {"enabled": False},
                    "iteration_based_train": {
                        "enabled": True, "iterations": args.iterations,
                        "save_period": max(args.iterations - 1, 1),
                        "train_log_step": 50, "valid_log_step": 50,
                        "valid_step": 10 ** 9, "lr_change_rate": 200},
                    "monitor": "off", "tensorboard": False,
                    "vis": {"enabled": False}},
        "train_dataloader": dl,
        "valid_dataloader": None,
    }

    device = torch.device(args.device)
    parser = ConfigParser(cfg, run_id="q0")
    logger = setup_logging("quality", parser.log_dir)
    trainer = build_training(parser, device, logger)
    trainer.train()

    ckpts = sorted(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))
    model = trainer.model.module if hasattr(trainer.model, "module") \
        else trainer.model
    model.eval()

    eval_ds = dict(ds)
    eval_ds["data_augment"] = {"enabled": False, "augment": [],
                               "augment_prob": []}
    dl_cfg = {"batch_size": 1, "shuffle": False, "num_workers": 0,
              "pin_memory": False, "drop_last": False, "use_ddp": False,
              "dataset": eval_ds}
    metrics = build_metrics(device)
    from esr_amd.data import read_datalist
    res = infer_sequence(dl_cfg, read_datalist(datalist)[0], model, device,
                         output_path=out / "eval", metrics=metrics,
                         save_images=True, max_batches=16)
    summary = {
        "iterations": args.iterations,
        "train_loss_avg": trainer.train_metrics.avg("train_loss"),
        "esr_rmse": res["esr_rmse"], "bicubic_rmse": res["bicubic_rmse"],
        "esr_mse": res["esr_mse"], "bicubic_mse": res["bicubic_mse"],
        "esr_l1": res["esr_l1"], "bicubic_l1": res["bicubic_l1"],
        "esr_beats_bicubic_rmse": bool(res["esr_rmse"] < res["bicubic_rmse"]),
        "checkpoint": str(ckpts[-1]) if ckpts else None,
    }
    with open(out / "summary.yml", "w") as f:
        yaml.safe_dump(summary, f)
    print(json.dumps(summary))


if __name__ == "__main__":
    main()
