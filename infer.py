#!/usr/bin/env python3
"""Inference / evaluation entry point (parity: ESR:infer_ours_cnt.py:135-350,
working mode 1: one model over a datalist of sequences)."""

import argparse
from pathlib import Path

import torch
import yaml

from esr_amd.data import read_datalist
from esr_amd.engine import load_model_from_checkpoint
from esr_amd.engine.inference import build_metrics, infer_sequence


def default_dataloader_config(args):
    """Inference defaults (parity: ESR:infer_ours_cnt.py:162-257)."""
    cfg = {
        "batch_size": 1, "shuffle": False, "num_workers": args.num_workers,
        "pin_memory": True, "drop_last": False, "use_ddp": False,
        "dataset": {
            "scale": args.scale, "ori_scale": args.ori_scale,
            "time_bins": args.time_bins, "mode": args.mode,
            "window": args.window, "sliding_window": args.sliding_window,
            "need_gt_frame": args.need_gt_frame,
            "need_gt_events": args.need_gt_events,
            "real_world_test": args.real_world_test,
            "data_augment": {"enabled": False, "augment": [], "augment_prob": []},
            "hot_filter": {"enabled": False},
            "sequence": {"sequence_length": args.seql, "seqn": args.seqn,
                         "step_size": args.step_size,
                         "pause": {"enabled": False,
                                   "proba_pause_when_running": 0.05,
                                   "proba_pause_when_paused": 0.9}},
        },
    }
    return cfg


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model_path", type=str, default=None)
    p.add_argument("--model_list", type=str, default=None,
                   help="txt of checkpoint paths; evaluates each")
    p.add_argument("--data_path", type=str, default=None)
    p.add_argument("--data_list", type=str, default=None)
    p.add_argument("--device", type=str, default="cuda:0")
    p.add_argument("--output_path", type=str, required=True)
    p.add_argument("--scale", type=int, default=2)
    p.add_argument("--seqn", type=int, default=3)
    p.add_argument("--seql", type=int, default=9)
    p.add_argument("--step_size", type=int, default=None)
    p.add_argument("--time_bins", type=int, default=1)
    p.add_argument("--ori_scale", type=str, default="down16")
    p.add_argument("--mode", type=str, default="events")
    p.add_argument("--window", type=int, default=2048)
    p.add_argument("--sliding_window", type=int, default=1024)
    p.add_argument("--need_gt_frame", action="store_true")
    p.add_argument("--need_gt_events", action="store_true")
    p.add_argument("--real_world_test", action="store_true")
    p.add_argument("--no_images", action="store_true")
    p.add_argument("--num_workers", type=int, default=2)
    p.add_argument("--lpips_weights", type=str, default=None,
                   help="linear-head state dict (defaults to the bundled "
                        "reference v0.1 heads)")
    p.add_argument("--lpips_backbone", type=str, default=None,
                   help="torchvision alexnet/vgg16 checkpoint for "
                        "paper-comparable LPIPS")
    args = p.parse_args()

    device = torch.device(args.device if torch.cuda.is_available() else "cpu")
    dl_cfg = default_dataloader_config(args)
    metrics = build_metrics(device, lpips_weights=args.lpips_weights,
                            lpips_backbone=args.lpips_backbone)

    if args.model_list:
        model_paths = read_datalist(args.model_list)
    elif args.model_path:
        model_paths = [args.model_path]
    else:
        raise SystemExit("provide --model_path or --model_list")

    if args.data_list:
        paths = read_datalist(args.data_list)
    elif args.data_path:
        paths = [args.data_path]
    else:
        raise SystemExit("provide --data_path or --data_list")

    out_root = Path(args.output_path)
    out_root.mkdir(parents=True, exist_ok=True)
    for model_path in model_paths:
        model, _ = load_model_from_checkpoint(model_path, device=device,
                                              seqn=args.seqn)
        mdir = out_root if len(model_paths) == 1 \
            else out_root / Path(model_path).stem
        all_results = {}
        for path in paths:
            name = Path(path).stem
            res = infer_sequence(dl_cfg, path, model, device,
                                 output_path=mdir / name, metrics=metrics,
                                 save_images=not args.no_images)
            all_results[name] = res
            print(f"{Path(model_path).stem} / {name}: {res}")

        mean = {}
        if all_results:
            first = next(iter(all_results.values()))
            for k, v in first.items():
                if isinstance(v, str):   # annotations (e.g. lpips_note)
                    mean[k] = v
                else:
                    mean[k] = sum(r[k] for r in all_results.values()) \
                        / len(all_results)
        mdir.mkdir(parents=True, exist_ok=True)
        with open(mdir / "mean_results.yml", "w") as f:
            yaml.safe_dump({"per_file": all_results, "mean": mean}, f)
        print("mean:", mean)


if __name__ == "__main__":
    main()
