#!/usr/bin/env python3
"""Plot training scalars from a workspace's scalars.jsonl (the JSONL
summary writer fallback; see mine_amd/utils/summary.py).

    python tools/plot_scalars.py /ws/v1 [--tags loss/train psnr_tgt/val]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
from collections import defaultdict


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("workspace")
    p.add_argument("--tags", nargs="*", default=None)
    p.add_argument("--out", default=None, help="output PNG (default: <ws>/scalars.png)")
    args = p.parse_args()

    path = os.path.join(args.workspace, "scalars.jsonl")
    series = defaultdict(lambda: ([], []))
    with open(path) as f:
        for line in f:
            d = json.loads(line)
            if args.tags and d["tag"] not in args.tags:
                continue
            xs, ys = series[d["tag"]]
            xs.append(d["step"])
            ys.append(d["value"])
    if not series:
        print("no matching scalars")
        return 1

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    n = len(series)
    fig, axes = plt.subplots((n + 2) // 3, min(n, 3),
                             figsize=(5 * min(n, 3), 3.2 * ((n + 2) // 3)),
                             squeeze=False)
    for ax, (tag, (xs, ys)) in zip(axes.flat, sorted(series.items())):
        ax.plot(xs, ys, lw=0.8)
        ax.set_title(tag, fontsize=9)
        ax.grid(alpha=0.3)
    for ax in list(axes.flat)[n:]:
        ax.axis("off")
    out = args.out or os.path.join(args.workspace, "scalars.png")
    fig.tight_layout()
    fig.savefig(out, dpi=110)
    print(f"wrote {out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
