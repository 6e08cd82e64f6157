#!/usr/bin/env python3
"""Plot objective-vs-time curves from driver stdout logs — the reference's
figure workflow (its README's experiments all end in error-vs-time plots of
the final ``time_ms,objective`` CSV block).

    python -m asyncframework_amd.cli asgd-thread ... > async.log
    python -m asyncframework_amd.cli asgd-sync  ... > sync.log
    python tools/plot_loss.py async.log sync.log -o fig.png

Each input may be a driver log (the CSV block after the last ``*********``
separator is used, exactly like the reference's plotted output) or a bare
``time_ms,objective`` CSV file.
"""

from __future__ import annotations

import argparse
import os
import re
import sys


def parse_curve(path: str):
    lines = open(path).read().splitlines()
    seps = [i for i, l in enumerate(lines) if l.startswith("*********")]
    start = seps[-1] + 1 if seps else 0
    pts = []
    for l in lines[start:]:
        if re.match(r"^\d+,[0-9.eE+-]+$", l):
            t, o = l.split(",")
            pts.append((int(t), float(o)))
    return pts


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("logs", nargs="+", help="driver logs or time,obj CSVs")
    ap.add_argument("-o", "--out", default="loss_curves.png")
    ap.add_argument("--logy", action="store_true", default=True)
    ap.add_argument("--title", default="objective vs wall-clock")
    args = ap.parse_args()

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(7, 4.5))
    any_pts = False
    for path in args.logs:
        pts = parse_curve(path)
        if not pts:
            print(f"[plot_loss] no curve in {path}", file=sys.stderr)
            continue
        any_pts = True
        xs = [t / 1000.0 for t, _ in pts]
        ys = [o for _, o in pts]
        ax.plot(xs, ys, marker="o", markersize=3,
                label=os.path.basename(path))
    if not any_pts:
        print("[plot_loss] nothing to plot", file=sys.stderr)
        sys.exit(1)
    if args.logy:
        ax.set_yscale("log")
    ax.set_xlabel("wall-clock (s)")
    ax.set_ylabel("objective")
    ax.set_title(args.title)
    ax.grid(True, alpha=0.3)
    ax.legend()
    fig.tight_layout()
    fig.savefig(args.out, dpi=120)
    print(f"[plot_loss] wrote {args.out}")


if __name__ == "__main__":
    main()
