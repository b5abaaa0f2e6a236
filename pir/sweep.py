"""Batch-PIR config sweep driver (the reference's sweep/sweep.py analog):
grid over hot fraction x collocation x bins x query budgets, one JSON
result per config, parallel over a process pool.

Usage: python -m pir.sweep --dataset lm|movielens|taobao --out DIR
         [--quick] [--processes 8]
"""

import argparse
import itertools
import json
import os
from multiprocessing import Pool

GRID = {
    "hot_fraction": [0.0, 0.01, 0.05, 0.1, 0.25],
    "group_size": [1, 2, 4],
    "num_bins": [8, 16, 32, 64],
    "queries_per_bin": [1, 2],
}

QUICK_GRID = {
    "hot_fraction": [0.0, 0.1],
    "group_size": [1, 2],
    "num_bins": [8, 32],
    "queries_per_bin": [1],
}


def _make_dataset(name, size, data_path=None):
    """size: quick (smoke), medium (full grid in minutes), full.
    data_path: real dataset location (WikiText-2-style dir / Taobao CSV
    dir / ml-20m ratings.csv); synthetic proxies when absent."""
    if name == "lm":
        from pir.datasets import language_model

        kw = {"quick": dict(vocab=512, corpus_len=20000),
              "medium": dict(vocab=1024, corpus_len=60000),
              "full": {}}[size]
        if data_path:
            kw = dict(data_path=data_path)
        ds = language_model.initialize(**kw)
        mb = {"quick": 20, "medium": 80, "full": None}[size]
        ds.train_model(epochs=1, max_batches=mb)
    elif name == "movielens":
        from pir.datasets import movielens

        kw = {"quick": dict(num_items=512, num_users=400),
              "medium": dict(num_items=2048, num_users=1500),
              "full": {}}[size]
        if data_path:
            kw = dict(data_path=data_path)
        ds = movielens.initialize(**kw)
        ds.train_model(epochs=1 if size == "quick" else 2)
    elif name == "taobao":
        from pir.datasets import taobao

        kw = {"quick": dict(num_items=512, num_samples=500),
              "medium": dict(num_items=4096, num_samples=2000),
              "full": {}}[size]
        if data_path:
            kw = dict(data_path=data_path,
                      num_samples={"quick": 2000, "medium": 20000,
                                   "full": None}[size])
        ds = taobao.initialize(**kw)
        ds.train_model(epochs=1 if size == "quick" else 2)
    else:
        raise ValueError(name)
    return ds


_DS = None


def _init_worker(name, size, data_path=None):
    global _DS
    import torch

    torch.manual_seed(0)
    _DS = _make_dataset(name, size, data_path)


def run_config(args):
    cfg, out_dir = args
    from pir import (BatchPIROptimize, CollocateConfig, HotColdConfig,
                     PIRConfig)

    ds = _DS
    opt = BatchPIROptimize(
        ds.num_entries, ds.train_patterns,
        hotcold=HotColdConfig(hot_fraction=cfg["hot_fraction"]),
        collocate=CollocateConfig(group_size=cfg["group_size"]),
        pir=PIRConfig(num_bins=cfg["num_bins"],
                      queries_per_bin=cfg["queries_per_bin"]),
    )
    result = dict(cfg)
    result.update(opt.summarize())
    result.update({"recovery": opt.evaluate(ds.eval_patterns)})
    result.update({"accuracy": opt.evaluate_real(ds)})
    name = "hf%s_g%s_b%s_q%s.json" % (cfg["hot_fraction"], cfg["group_size"],
                                      cfg["num_bins"], cfg["queries_per_bin"])
    with open(os.path.join(out_dir, name), "w") as f:
        json.dump(result, f, indent=1)
    return result


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="lm",
                    choices=["lm", "movielens", "taobao"])
    ap.add_argument("--out", default="pir/sweep_out")
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--size", default=None,
                    choices=["quick", "medium", "full"])
    ap.add_argument("--processes", type=int, default=8)
    ap.add_argument("--data-path", default=None,
                    help="real dataset location (lm: WikiText-2 dir or "
                         ".txt; taobao: CSV dir; movielens: ratings.csv)")
    a = ap.parse_args()
    size = a.size or ("quick" if a.quick else "full")
    grid = QUICK_GRID if size == "quick" else GRID
    out_dir = os.path.join(a.out, a.dataset)
    os.makedirs(out_dir, exist_ok=True)
    keys = list(grid)
    cfgs = [dict(zip(keys, vals)) for vals in
            itertools.product(*(grid[k] for k in keys))]
    if a.processes <= 1:
        _init_worker(a.dataset, size, a.data_path)
        results = [run_config((c, out_dir)) for c in cfgs]
    else:
        with Pool(a.processes, initializer=_init_worker,
                  initargs=(a.dataset, size, a.data_path)) as pool:
            results = pool.map(run_config, [(c, out_dir) for c in cfgs])
    print("wrote %d configs to %s" % (len(results), out_dir))


if __name__ == "__main__":
    main()
