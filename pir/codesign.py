"""Join batch-PIR accuracy sweeps with GPU kernel perf sweeps
(the reference's codesign/join_batch_pir_accuracy_with_gpu_dpf.py analog).

For each (accuracy-sweep config, kernel-perf measurement) pair whose table
shapes are compatible, computes the serving-side operating point assuming
a 2-GPU deployment (one per PIR server):

    queries/batch   = num_bins x queries_per_bin
    latency         = kernel latency at bin-sized tables (batch = queries)
    throughput      = kernel DPFs/sec / queries_per_batch  (batches/sec)

and emits accuracy-vs-throughput points under latency/communication
constraints, with Pareto extraction.

Usage: python -m pir.codesign --accuracy pir/sweep_out/lm \
          --perf benchmarks/sweep_out/sweep.csv --out pir/codesign_lm.json
"""

import argparse
import csv
import glob
import json
import os

from pir.batch_pir import pareto_front


def load_accuracy(dir_path):
    out = []
    for p in sorted(glob.glob(os.path.join(dir_path, "*.json"))):
        with open(p) as f:
            out.append(json.load(f))
    return out


def _num(v):
    try:
        return float(v)
    except (TypeError, ValueError):
        return v


def load_perf(csv_path):
    """GPU kernel-sweep rows with numeric throughput and latency
    (CPU-baseline rows in mixed CSVs are skipped)."""
    with open(csv_path) as f:
        rows = [{k: _num(v) for k, v in row.items()}
                for row in csv.DictReader(f)]
    return [r for r in rows
            if isinstance(r.get("throughput_dpfs_per_sec"), float)
            and isinstance(r.get("latency_ms"), float)
            and isinstance(r.get("num_entries"), float)]


def join(acc_rows, perf_rows, max_latency_ms=100.0, max_comm_bytes=300_000,
         num_gpus=2):
    points = []
    for a in acc_rows:
        if a.get("communication_bytes", 0) > max_comm_bytes:
            continue
        queries = a["num_bins"] * a["queries_per_bin"]
        bin_entries = max(128, a["num_entries"] // max(1, a["num_bins"]))
        # best perf row whose table is >= the bin size
        cands = [p for p in perf_rows
                 if p.get("num_entries", 0) >= bin_entries]
        if not cands:
            continue
        p = min(cands, key=lambda r: r["num_entries"])
        latency = p.get("latency_ms", 0.0) + 1.0  # + network budget
        if latency > max_latency_ms:
            continue
        dpfs = p["throughput_dpfs_per_sec"]
        batches_per_sec = dpfs * num_gpus / (2 * queries)  # 2 servers
        points.append({
            "config": {k: a[k] for k in ("hot_fraction", "group_size",
                                         "num_bins", "queries_per_bin")},
            "accuracy": a["accuracy"],
            "communication_bytes": a["communication_bytes"],
            "latency_ms": latency,
            "batches_per_sec": batches_per_sec,
            "perf_row": {k: p.get(k) for k in ("num_entries", "batch_size",
                                               "prf", "strategy")},
        })
    metric_sign = 1.0
    if points and points[0]["accuracy"].get("metric") == "ppl":
        metric_sign = -1.0  # lower ppl is better
    xy = [(-pt["batches_per_sec"], metric_sign * pt["accuracy"]["value"])
          for pt in points]
    front = set(pareto_front(xy)) if points else set()
    for i, pt in enumerate(points):
        pt["pareto"] = i in front
    return points


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--accuracy", required=True)
    ap.add_argument("--perf", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--max-latency-ms", type=float, default=100.0)
    ap.add_argument("--max-comm-bytes", type=int, default=300_000)
    a = ap.parse_args()
    pts = join(load_accuracy(a.accuracy), load_perf(a.perf),
               a.max_latency_ms, a.max_comm_bytes)
    with open(a.out, "w") as f:
        json.dump(pts, f, indent=1)
    print("wrote %d operating points (%d on the Pareto front) to %s"
          % (len(pts), sum(p["pareto"] for p in pts), a.out))


if __name__ == "__main__":
    main()
