#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel-stats CSV: top kernels by total time +
bucketed shares (native / MIOpen-conv / transpose / cast / elementwise).

Usage: python tools/summarize_profile.py <stats.csv> [topN]
"""

import csv
import re
import sys


BUCKETS = [
    ("native-conv", re.compile(r"conv2d_fwd|conv2d_dgrad|conv2d_wgrad")),
    ("native-other", re.compile(r"dcn_|gru_gates|splat_|redist_|act_grad")),
    ("miopen-conv", re.compile(r"igemm|naive_conv|miopen|Winograd|Conv",
                               re.IGNORECASE)),
    ("transpose", re.compile(r"batched_transpose|Transpose")),
    ("cast/copy", re.compile(r"cast|copy_|direct_copy|elementwise_kernel.*"
                             r"(BFloat16.*float|float.*BFloat16)")),
    ("gemm", re.compile(r"Cijk|gemm", re.IGNORECASE)),
    ("rccl", re.compile(r"nccl|rccl", re.IGNORECASE)),
]


def bucket(name):
    for label, pat in BUCKETS:
        if pat.search(name):
            return label
    return "other"


def main():
    path = sys.argv[1]
    topn = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    rows = list(csv.DictReader(open(path)))
    dur = "TotalDurationNs" if "TotalDurationNs" in rows[0] else "DurationNs"
    calls_key = "Calls" if "Calls" in rows[0] else "Count"
    rows.sort(key=lambda r: -float(r[dur]))
    tot = sum(float(r[dur]) for r in rows)
    print(f"total {tot / 1e6:.1f} ms over {len(rows)} distinct kernels")

    shares = {}
    for r in rows:
        b = bucket(r["Name"])
        shares[b] = shares.get(b, 0.0) + float(r[dur])
    print("\nbuckets:")
    for b, v in sorted(shares.items(), key=lambda kv: -kv[1]):
        print(f"  {b:14s} {v / 1e6:9.2f} ms  {100 * v / tot:5.1f}%")

    print(f"\ntop {topn} kernels:")
    for r in rows[:topn]:
        print(f'{float(r[dur]) / 1e6:9.2f}ms {100 * float(r[dur]) / tot:5.1f}% '
              f'n={r[calls_key]:>6s} [{bucket(r["Name"]):12s}] '
              f'{r["Name"][:90]}')


if __name__ == "__main__":
    main()
