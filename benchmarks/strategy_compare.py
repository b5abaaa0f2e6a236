"""Strategy comparison at a fixed config (dict-line per strategy):
fused (production), two_stage (expand + MFMA matmul), expand (one-hot
only), naive (O(n log n) oracle)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse
import time

import torch

from gpudpf import DPF


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=65536)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--prf", default="AES128")
    ap.add_argument("--reps", type=int, default=8)
    a = ap.parse_args()
    prf = getattr(DPF, "PRF_" + a.prf)
    dpf = DPF(prf=prf)
    k1, _ = dpf.gen(1, a.n)
    keys = torch.stack([k1] * a.batch)
    table = torch.randint(-(2**31), 2**31 - 1, (a.n, 16), dtype=torch.int64).to(
        torch.int32)
    dpf.eval_init(table)

    def timed(fn):
        t0 = time.time()
        while time.time() - t0 < 0.4:
            fn()
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(a.reps):
            fn()
        torch.cuda.synchronize()
        return (time.time() - t0) / a.reps * 1e3

    rows = {
        "fused": lambda: dpf.eval_gpu(keys),
        "two_stage": lambda: dpf.eval_gpu(keys, strategy="two_stage"),
        "expand(one-hot)": lambda: dpf.eval_gpu(keys, one_hot_only=True),
        "bfs(one-hot)": lambda: dpf.eval_gpu(keys, one_hot_only=True,
                                             strategy="bfs"),
    }
    # single-key latency strategies (batch=1): j-split fused vs the
    # grid-synchronized cooperative kernel
    one = keys[:1]
    rows["fused(batch=1)"] = lambda: dpf.eval_gpu(one)
    rows["coop(batch=1)"] = lambda: dpf.eval_gpu(one, strategy="coop")
    for name, fn in rows.items():
        ms = timed(fn)
        nb = 1 if "batch=1" in name else a.batch
        print({"strategy": name, "prf": a.prf, "n": a.n, "batch": nb,
               "ms_per_batch": round(ms, 3),
               "dpfs_per_sec": round(nb / ms * 1e3, 1)})


if __name__ == "__main__":
    main()
