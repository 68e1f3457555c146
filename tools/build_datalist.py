#!/usr/bin/env python3
"""Build train/valid datalist txt files from a directory of EVS sequences
(parity: ESR:datalist/generate_datalist.py)."""

import argparse
from pathlib import Path


def main():
    p = argparse.ArgumentParser()
    p.add_argument("root", help="directory containing *.evs sequence dirs")
    p.add_argument("--out", default=None)
    p.add_argument("--split", type=float, default=0.8)
    args = p.parse_args()

    root = Path(args.root)
    out = Path(args.out or root)
    seqs = sorted(str(d) for d in root.glob("*.evs")
                  if (d / "meta.json").exists())
    if not seqs:
        raise SystemExit(f"no EVS sequences under {root}")
    n_train = max(int(len(seqs) * args.split), 1)
    (out / "train_datalist.txt").write_text("\n".join(seqs[:n_train]) + "\n")
    (out / "valid_datalist.txt").write_text(
        "\n".join(seqs[n_train:] or seqs[-1:]) + "\n")
    print(f"{n_train} train / {len(seqs) - n_train} valid sequences")


if __name__ == "__main__":
    main()
