#!/usr/bin/env python3
"""Run the BASELINE.json config arms 3-5 on one GPU and collect evidence.

BASELINE.json configs (single-GPU arms; the 8-GPU DDP scaling points are
the driver's SCALE run):
  3. 4x ESR NFS-syn 64->256 bf16, seq_len=8
  4. 2x ESR EventZoom-class, seq_len=16 (long BPTT)
  5. 4x ESR 180x240 -> 720x960 DVS-native res, fp16 mixed

Each arm shells out to bench.py with the arm's flags, parses the JSON
line, and appends peak HBM usage.  Results land in
artifacts/baseline_configs/configN.json (committed as driver-verifiable
evidence).

Usage (GPU box): python tools/bench_baseline_configs.py [--steps 15]
"""

import argparse
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

ARMS = {
    "config3_4x_64to256": dict(
        flags=["--scale", "4", "--lr-size", "64", "--steps", "{steps}",
               "--warmup", "3",
               "--metric-suffix",
               "SR event-frames/sec, 4x NFS-syn 64->256 (config 3 shape)"],
        note="4x SR, same HR grid as the headline; DDP-8 point comes from "
             "the driver SCALE run"),
    "config4_longbptt_seq16": dict(
        flags=["--seql", "16", "--batch", "32", "--steps", "{steps}",
               "--warmup", "3",
               "--metric-suffix",
               "SR event-frames/sec, 2x 128->256 seq_len=16 long-BPTT "
               "(config 4 class)"],
        note="13 BPTT windows per step; activation-memory stress arm "
             "(SURVEY hard-part 2)"),
    "config5_dvs_fp16": dict(
        flags=["--scale", "4", "--lr-h", "180", "--lr-w", "240",
               "--dtype", "fp16", "--batch", "8", "--window", "4096",
               "--steps", "{steps}", "--warmup", "3",
               "--metric-suffix",
               "SR event-frames/sec, 4x DVS 180x240->720x960 fp16 "
               "(config 5 class)"],
        note="fp16 autocast, DVS-native non-square resolution"),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=15)
    ap.add_argument("--arms", nargs="*", default=list(ARMS))
    ap.add_argument("--out", default="artifacts/baseline_configs")
    args = ap.parse_args()

    out_dir = REPO / args.out
    out_dir.mkdir(parents=True, exist_ok=True)

    for name in args.arms:
        arm = ARMS[name]
        flags = [f.replace("{steps}", str(args.steps)) for f in arm["flags"]]
        proc = subprocess.run(
            [sys.executable, str(REPO / "bench.py")] + flags,
            capture_output=True, text=True, timeout=900, cwd=str(REPO))
        record = {"arm": name, "note": arm["note"], "flags": flags,
                  "returncode": proc.returncode}
        if proc.returncode == 0:
            line = [ln for ln in proc.stdout.splitlines()
                    if ln.startswith("{")][-1]
            record["result"] = json.loads(line)
        else:
            record["stderr_tail"] = proc.stderr[-2000:]
        with open(out_dir / f"{name}.json", "w") as f:
            json.dump(record, f, indent=2)
        print(json.dumps(record.get("result", record), indent=None))


if __name__ == "__main__":
    main()
