"""Inference / evaluation harness.

Parity: ESR:infer_ours_cnt.py:22-115 (`infer_body`): per-sequence recurrent
inference, ESR-vs-bicubic metric tables (L1 / MSE / LPIPS / SSIM / PSNR),
forward-pass timing, per-file + aggregate YAML results, PNG renders of
lr / scaled / bicubic / esr / gt count maps.

On GPU the forward is timed with hipEvents (torch.cuda.Event) instead of
the reference's host wall-clock.
"""

from __future__ import annotations

import time
from pathlib import Path

import torch
import torch.nn.functional as F
import yaml

from ..data import InferenceSequenceDataLoader
from ..loss import PerceptualLoss, l1 as l1_fn, mse as mse_fn, \
    psnr as psnr_fn, rmse as rmse_fn, ssim as ssim_fn
from ..utils import MetricTracker
from ..utils.vis import EventVisualizer

__all__ = ["infer_sequence", "build_metrics"]

_IMG_DIRS = ["lr_event_img", "hr_scaled_event_img", "hr_esr_event_img",
             "hr_bicubic_event_img", "hr_gt_event_img"]


def build_metrics(device="cpu", lpips_net="alex", lpips_weights=None,
                  lpips_backbone=None):
    return {
        "l1": l1_fn,
        "mse": mse_fn,
        "ssim": ssim_fn,
        "psnr": psnr_fn,
        "lpips": PerceptualLoss(net=lpips_net, device=device,
                                weights_path=lpips_weights,
                                backbone_path=lpips_backbone),
    }


@torch.no_grad()
def infer_sequence(dataloader_config, data_path, model, device,
                   output_path=None, metrics=None, save_images=True,
                   max_batches=None):
    metrics = metrics or build_metrics(device)
    vis = EventVisualizer()
    dataloader = InferenceSequenceDataLoader(data_path, dataloader_config)
    gt_res = dataloader.gt_sensor_resolution

    keys = ["esr_l1", "esr_mse", "esr_rmse", "esr_lpips", "esr_ssim",
            "esr_psnr", "bicubic_l1", "bicubic_mse", "bicubic_rmse",
            "bicubic_lpips", "bicubic_ssim", "bicubic_psnr", "time", "params"]
    track = MetricTracker(keys)

    if output_path is not None:
        event_img_path = Path(output_path) / "event_imgs"
        img_path = Path(output_path) / "imgs"
        for d in _IMG_DIRS:
            (event_img_path / d).mkdir(parents=True, exist_ok=True)
        (img_path / "gt_img").mkdir(parents=True, exist_ok=True)

    seq_cfg = dataloader_config["dataset"]["sequence"]
    mid_idx = (seq_cfg["seqn"] - 1) // 2

    model.eval()
    if hasattr(model, "reset_states"):
        model.reset_states()
    use_events = device.type == "cuda" if isinstance(device, torch.device) \
        else str(device).startswith("cuda")

    for i, inputs_seq in enumerate(dataloader):
        if max_batches is not None and i >= max_batches:
            break
        inputs = inputs_seq[0]
        inp_cnt = inputs["inp_cnt"][:, mid_idx]
        inp_scaled_cnt = inputs["inp_scaled_cnt"].to(device)
        gt_cnt = inputs["gt_cnt"][:, mid_idx]

        if i == 0:
            params = sum(p.numel() for p in model.parameters())
            track.update("params", params / 1e6)

        if use_events:
            ev0 = torch.cuda.Event(enable_timing=True)
            ev1 = torch.cuda.Event(enable_timing=True)
            ev0.record()
            esr_cnt = model(inp_scaled_cnt)
            ev1.record()
            ev1.synchronize()
            total_time = ev0.elapsed_time(ev1) / 1e3
        else:
            t0 = time.time()
            esr_cnt = model(inp_scaled_cnt)
            total_time = time.time() - t0
        esr_cnt = esr_cnt.float().cpu()
        if esr_cnt.shape[-2:] != gt_cnt.shape[-2:]:
            esr_cnt = F.interpolate(esr_cnt, size=gt_cnt.shape[-2:],
                                    mode="bicubic", align_corners=False)
        bicubic_cnt = F.interpolate(inp_cnt, size=gt_res, mode="bicubic",
                                    align_corners=False)

        for name, pred in (("esr", esr_cnt), ("bicubic", bicubic_cnt)):
            track.update(f"{name}_l1", metrics["l1"](pred, gt_cnt).item())
            track.update(f"{name}_mse", metrics["mse"](pred, gt_cnt).item())
            track.update(f"{name}_rmse", rmse_fn(pred, gt_cnt).item())
            track.update(f"{name}_ssim", metrics["ssim"](pred, gt_cnt))
            track.update(f"{name}_psnr", metrics["psnr"](pred, gt_cnt))
            track.update(f"{name}_lpips",
                         metrics["lpips"](pred.to(device), gt_cnt.to(device))
                         .cpu().item())
        track.update("time", total_time)

        if save_images and output_path is not None:
            def _p(d):
                return str(event_img_path / d / f"{i:09d}.png")
            hwc = lambda t: t.cpu().numpy().transpose(1, 2, 0)  # noqa: E731
            vis.plot_event_cnt(hwc(inputs["inp_cnt"][0, mid_idx]), True, _p("lr_event_img"))
            vis.plot_event_cnt(hwc(inputs["inp_scaled_cnt"][0, mid_idx]), True, _p("hr_scaled_event_img"))
            vis.plot_event_cnt(hwc(bicubic_cnt[0]), True, _p("hr_bicubic_event_img"))
            vis.plot_event_cnt(hwc(esr_cnt[0].round()), True, _p("hr_esr_event_img"))
            vis.plot_event_cnt(hwc(inputs["gt_cnt"][0, mid_idx]), True, _p("hr_gt_event_img"))
            vis.plot_frame((inputs["gt_img"][0, mid_idx, 0].numpy() * 255).astype("uint8"),
                           True, str(img_path / "gt_img" / f"{i:09d}.png"))

    result = track.result()
    lp = metrics.get("lpips")
    if lp is not None and hasattr(lp, "paper_comparable") \
            and not lp.paper_comparable:
        # trained linear heads ship in-repo, but without an ImageNet backbone
        # checkpoint the feature extractor is a seeded random projection —
        # flag it so lpips numbers are not compared against paper tables
        result["lpips_note"] = ("random-init backbone (no ImageNet checkpoint"
                                " in env): lpips values are relative-only")
    if output_path is not None:
        with open(Path(output_path) / "results.yml", "w") as f:
            yaml.safe_dump({"evaluation results": result}, f)
    return result
