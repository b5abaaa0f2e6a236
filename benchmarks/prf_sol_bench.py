"""PRF speed-of-light microbenchmark: dependent chains of pair expansions
on register data only — the upper bound for any expansion kernel.  The
fused kernel's pair rate divided by this is its efficiency."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from gpudpf import _hip

NAMES = {0: "DUMMY", 1: "SALSA20", 2: "CHACHA20", 3: "AES128"}


def main(blocks=2048, iters=2000):
    aes_ptr = _hip.ensure_aes_tables(0)
    stream = torch.cuda.current_stream().cuda_stream
    out = torch.empty(blocks * 256, dtype=torch.int32, device="cuda:0")
    results = {}
    for prf in (1, 2, 3, 0):
        _hip.prf_sol(aes_ptr, out.data_ptr(), blocks, 100, prf, stream)
        torch.cuda.synchronize()
        t0 = time.time()
        reps = 3
        for _ in range(reps):
            _hip.prf_sol(aes_ptr, out.data_ptr(), blocks, iters, prf, stream)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / reps
        pairs = blocks * 256 * iters
        results[NAMES[prf]] = pairs / dt / 1e9
        print("%-8s SOL: %8.2f Gpair/s  (%.2f ms per launch)"
              % (NAMES[prf], pairs / dt / 1e9, dt * 1e3))
    # fused-kernel effective pair rates at n=2^20 batch 512 for comparison
    for name, ms in (("SALSA20", 14.6), ("CHACHA20", 15.6), ("AES128", 19.8)):
        rate = 512 * (1 << 20) / (ms * 1e-3) / 1e9
        print("%-8s fused: %6.2f Gpair/s -> %.0f%% of SOL"
              % (name, rate, 100 * rate / results[name]))


if __name__ == "__main__":
    main()
