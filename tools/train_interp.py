#!/usr/bin/env python3
"""Self-train the frame interpolator on synthetic motion sequences.

The reference downloads a pretrained Super-SloMo checkpoint for its
dataset-generation pipeline; this environment has no network, so the
FrameInterpolator ships random-init.  This tool closes that gap: it
generates synthetic translating/rotating texture triplets (I0, It, I1),
trains the interpolator with photometric + warp losses, and saves a
checkpoint `upsample_frames` can load — making the offline generation
pipeline self-contained.

  python tools/train_interp.py --steps 300 --out runs/interp.pth
"""

import argparse
import math
import sys
from pathlib import Path

import torch
import torch.nn.functional as F

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def synth_triplet(batch, res, gen, device):
    """Random smooth textures translated by a random global motion:
    It is the ground-truth middle frame (t=0.5)."""
    H, W = res
    base = torch.rand(batch, 1, H * 2, W * 2, generator=gen, device=device)
    # lightly smooth (sharp enough that interpolation must learn flow)
    base = F.avg_pool2d(base, 3, 1, 1)
    dx = (torch.rand(batch, generator=gen, device=device) * 2 - 1) * 6
    dy = (torch.rand(batch, generator=gen, device=device) * 2 - 1) * 6

    def crop(shift):
        out = []
        for b in range(batch):
            ox = int(W // 2 + shift * dx[b])
            oy = int(H // 2 + shift * dy[b])
            out.append(base[b, :, oy:oy + H, ox:ox + W])
        return torch.stack(out)

    return crop(-0.5), crop(0.0), crop(0.5)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--res", type=int, default=64)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--out", default="runs/interp.pth")
    ap.add_argument("--device", default="cuda:0"
                    if torch.cuda.is_available() else "cpu")
    ap.add_argument("--log-every", type=int, default=50)
    args = ap.parse_args()

    from esr_amd.models.interp import FrameInterpolator, backwarp

    device = torch.device(args.device)
    torch.manual_seed(0)
    gen = torch.Generator(device=device).manual_seed(1)
    model = FrameInterpolator().to(device)
    opt = torch.optim.Adam(model.parameters(), lr=args.lr)

    first = last = None
    for step in range(args.steps):
        I0, It, I1 = synth_triplet(args.batch, (args.res, args.res), gen,
                                   device)
        pred, flows = model(I0, I1, 0.5, return_flows=True)
        f01, f10 = flows[0], flows[1]
        # reconstruction + bidirectional warp losses (Super-SloMo recipe)
        loss_r = F.l1_loss(pred, It)
        loss_w = F.l1_loss(backwarp(I0, f10), I1) + \
            F.l1_loss(backwarp(I1, f01), I0)
        # flow smoothness
        loss_s = (f01.diff(dim=-1).abs().mean() +
                  f01.diff(dim=-2).abs().mean() +
                  f10.diff(dim=-1).abs().mean() +
                  f10.diff(dim=-2).abs().mean())
        loss = loss_r + 0.5 * loss_w + 0.05 * loss_s
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
        last = loss.item()
        if step % args.log_every == 0:
            print(f"step {step}: loss {loss.item():.4f} "
                  f"(recon {loss_r.item():.4f})")

    out = Path(args.out)
    out.parent.mkdir(parents=True, exist_ok=True)
    torch.save({"model": {"name": "FrameInterpolator",
                          "states": model.state_dict()},
                "final_loss": last}, out)
    print(f"saved {out}: loss {first:.4f} -> {last:.4f}")
    return first, last


if __name__ == "__main__":
    main()
