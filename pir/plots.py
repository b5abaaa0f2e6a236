"""Pareto plots for sweeps and co-design joins (the reference's
sweep/*_plot.py + codesign/plot_*.py analog).

Usage:
  python -m pir.plots accuracy pir/sweep_out/lm --out lm_pareto.png
  python -m pir.plots codesign pir/codesign_lm.json --out lm_codesign.png
"""

import argparse
import glob
import json
import os

from pir.batch_pir import pareto_front


def _load_dir(d):
    rows = []
    for p in sorted(glob.glob(os.path.join(d, "*.json"))):
        with open(p) as f:
            rows.append(json.load(f))
    return rows


def plot_accuracy(dir_path, out, max_comm=300_000):
    rows = [r for r in _load_dir(dir_path)
            if r.get("communication_bytes", 0) <= max_comm]
    if not rows:
        raise SystemExit("no sweep results in " + dir_path)
    metric = rows[0]["accuracy"]["metric"]
    lower_better = metric == "ppl"
    xs = [r["communication_bytes"] / 1024 for r in rows]
    ys = [r["accuracy"]["value"] for r in rows]
    pts = [(x, -y if lower_better else y) for x, y in zip(xs, ys)]
    front = pareto_front(pts)
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    plt.figure(figsize=(6, 4))
    plt.scatter(xs, ys, s=14, alpha=0.5, label="configs")
    fx = [xs[i] for i in front]
    fy = [ys[i] for i in front]
    order = sorted(range(len(fx)), key=lambda i: fx[i])
    plt.plot([fx[i] for i in order], [fy[i] for i in order], "r-o",
             label="pareto")
    plt.xlabel("communication per batch (KB)")
    plt.ylabel(metric)
    plt.legend()
    plt.tight_layout()
    plt.savefig(out, dpi=120)
    print("wrote", out)


def plot_codesign(join_path, out):
    with open(join_path) as f:
        pts = json.load(f)
    if not pts:
        raise SystemExit("empty join file")
    metric = pts[0]["accuracy"]["metric"]
    xs = [p["batches_per_sec"] for p in pts]
    ys = [p["accuracy"]["value"] for p in pts]
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    plt.figure(figsize=(6, 4))
    plt.scatter(xs, ys, s=14, alpha=0.5)
    par = [(x, y) for p, x, y in zip(pts, xs, ys) if p.get("pareto")]
    if par:
        par.sort()
        plt.plot([x for x, _ in par], [y for _, y in par], "r-o",
                 label="pareto")
        plt.legend()
    plt.xscale("log")
    plt.xlabel("batches/sec (2 x MI355X)")
    plt.ylabel(metric)
    plt.tight_layout()
    plt.savefig(out, dpi=120)
    print("wrote", out)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("kind", choices=["accuracy", "codesign"])
    ap.add_argument("path")
    ap.add_argument("--out", required=True)
    a = ap.parse_args()
    if a.kind == "accuracy":
        plot_accuracy(a.path, a.out)
    else:
        plot_codesign(a.path, a.out)
