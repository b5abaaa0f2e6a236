"""CPU-baseline benchmark (the reference's paper/kernel/cpu/dpf_google
analog): multithreaded CPU DPF expansion + table inner product, printed as
a dict line.  Our CPU core expands with O(n) PRF pairs per key (the
reference's CPU path is O(n log n) single calls), so this baseline is
itself stronger than the original.

Usage: python benchmarks/cpu_benchmark.py [--n 16384] [--batch 512]
         [--threads 32] [--prf AES128] [--reps 3] [--use-matmul 1]
Thread sweep: bash benchmarks/cpu_thread_sweep.sh
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import time

import torch

from gpudpf import DPF

PRF_IDS = {
    "DUMMY": DPF.PRF_DUMMY,
    "SALSA20": DPF.PRF_SALSA20,
    "CHACHA20": DPF.PRF_CHACHA20,
    "AES128": DPF.PRF_AES128,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=16384)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--threads", type=int, default=32)
    ap.add_argument("--prf", default="AES128", choices=list(PRF_IDS))
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--entry-size", type=int, default=16)
    ap.add_argument("--use-matmul", type=int, default=1)
    a = ap.parse_args()

    prf = PRF_IDS[a.prf]
    dpf = DPF(prf=prf, device="cpu")
    k1, _ = dpf.gen(1, a.n)
    keys = [k1] * a.batch
    table = torch.randint(-(2**31), 2**31 - 1, (a.n, a.entry_size),
                          dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)

    t0 = time.time()
    for _ in range(a.reps):
        if a.use_matmul:
            dpf.eval_cpu(keys, num_threads=a.threads)
        else:
            dpf.eval_cpu(keys, one_hot_only=True, num_threads=a.threads)
    elapsed = time.time() - t0
    result = {
        "backend": "cpu",
        "prf": a.prf,
        "num_entries": a.n,
        "batch_size": a.batch,
        "entry_size": a.entry_size,
        "threads": a.threads,
        "reps": a.reps,
        "use_matmul": a.use_matmul,
        "throughput_dpfs_per_sec": round(a.batch * a.reps / elapsed, 1),
    }
    print(result)


if __name__ == "__main__":
    main()
