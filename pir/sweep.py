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


def _make_dataset(name, quick):
    if name == "lm":
        from pir.datasets import language_model

        kw = dict(vocab=512, corpus_len=20000) if quick else {}
        ds = language_model.initialize(**kw)
        ds.train_model(epochs=1, max_batches=20 if quick else None)
    elif name == "movielens":
        from pir.datasets import movielens

        kw = dict(num_items=512, num_users=400) if quick else {}
        ds = movielens.initialize(**kw)
        ds.train_model(epochs=1 if quick else 2)
    elif name == "taobao":
        from pir.datasets import taobao

        kw = dict(num_items=512, num_samples=500) if quick else {}
        ds = taobao.initialize(**kw)
        ds.train_model(epochs=1 if quick else 2)
    else:
        raise ValueError(name)
    return ds


_DS = None


def _init_worker(name, quick):
    global _DS
    import torch

    torch.manual_seed(0)
    _DS = _make_dataset(name, quick)


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
    ap.add_argument("--processes", type=int, default=8)
    a = ap.parse_args()
    grid = QUICK_GRID if a.quick else GRID
    out_dir = os.path.join(a.out, a.dataset)
    os.makedirs(out_dir, exist_ok=True)
    keys = list(grid)
    cfgs = [dict(zip(keys, vals)) for vals in
            itertools.product(*(grid[k] for k in keys))]
    if a.processes <= 1:
        _init_worker(a.dataset, a.quick)
        results = [run_config((c, out_dir)) for c in cfgs]
    else:
        with Pool(a.processes, initializer=_init_worker,
                  initargs=(a.dataset, a.quick)) as pool:
            results = pool.map(run_config, [(c, out_dir) for c in cfgs])
    print("wrote %d configs to %s" % (len(results), out_dir))


if __name__ == "__main__":
    main()
