"""Kernel research benchmark (the reference's dpf_benchmark.cu analog:
paper/kernel/gpu/dpf_benchmark.cu).

Measures, per (strategy, prf, n, batch, entry_size) config:
  - throughput (DPFs/sec) over REPS repetitions on the serving path
  - single-shot latency (ms) at batch=1 (via the j-split grid; the
    reference needs a separate cooperative kernel for this)
and prints a final python-dict line for the sweep scraper
(benchmarks/scrape.py), preserving the reference's printf-dict contract
(dpf_benchmark.cu:307-314 -> scripts/scrape.py).

Strategies:
  fused  - production per-thread-DFS expansion fused with the table MAC
  expand - full one-hot share expansion (breadth-first-equivalent output)
  naive  - O(n log n) per-leaf oracle kernel
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import time

import torch

from gpudpf import DPF, _core

try:
    from gpudpf import _hip
except ImportError:
    _hip = None

PRF_IDS = {
    "DUMMY": DPF.PRF_DUMMY,
    "SALSA20": DPF.PRF_SALSA20,
    "CHACHA20": DPF.PRF_CHACHA20,
    "AES128": DPF.PRF_AES128,
}


def run(strategy, prf_name, n, batch, entry_size, reps, check=False):
    assert torch.cuda.is_available(), "kernel benchmark needs a GPU"
    prf = PRF_IDS[prf_name]
    dpf = DPF(prf=prf)
    k1, k2 = dpf.gen(n // 2 + 1, n)
    keys_cpu = torch.stack([k1] * batch)
    table = torch.randint(-(2**31), 2**31 - 1, (n, entry_size), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    dev = torch.device(dpf.device)
    keys_gpu = keys_cpu.to(dev).contiguous()
    depth = n.bit_length() - 1
    stream = torch.cuda.current_stream(dev).cuda_stream
    aes_ptr = dpf._aes_ptr

    def launch(b, keys_g):
        if strategy == "fused":
            out = torch.zeros((b, 16), dtype=torch.int32, device=dev)
            _hip.eval_fused(keys_g.data_ptr(), dpf._table_gpu.data_ptr(),
                            out.data_ptr(), aes_ptr, b, n, depth, dpf._zlog,
                            prf, stream)
        elif strategy == "expand":
            out = torch.empty((b, n), dtype=torch.int32, device=dev)
            _hip.eval_expand(keys_g.data_ptr(), out.data_ptr(), aes_ptr, b, n,
                             depth, dpf._zlog, prf, stream)
        elif strategy == "naive":
            out = torch.empty((b, n), dtype=torch.int32, device=dev)
            _hip.eval_naive(keys_g.data_ptr(), out.data_ptr(), aes_ptr, b, n,
                            depth, prf, stream)
        elif strategy == "bfs":
            out = torch.empty((b, n), dtype=torch.int32, device=dev)
            _hip.eval_bfs(keys_g.data_ptr(), out.data_ptr(), aes_ptr, b, n,
                          depth, prf, stream)
        elif strategy == "coop":
            out = torch.zeros((b, 16), dtype=torch.int32, device=dev)
            for i in range(b):
                _hip.eval_coop(keys_g[i].data_ptr(),
                               dpf._table_gpu.data_ptr(), out[i].data_ptr(),
                               aes_ptr, n, depth, dpf._zlog, prf, True,
                               stream)
        else:
            raise ValueError(strategy)
        return out

    # correctness oracle (DUMMY-PRF full-output check, like the reference's
    # check_correct/check_correct_fused gated on the fake-crypto backend)
    if check:
        want_shares = torch.from_numpy(_core.expand(k1.numpy(), prf))
        if strategy == "fused":
            got = launch(1, keys_gpu[:1])[0].cpu()
            padded = torch.nn.functional.pad(table, (0, 16 - entry_size))
            want = torch.from_numpy(
                _core.eval_fused_cpu(k1.numpy(), padded.numpy(), prf))
            assert torch.equal(got, want), "fused check failed"
        elif strategy == "expand":
            got = launch(1, keys_gpu[:1])[0].cpu()
            perm = torch.from_numpy(_core.leaf_perm_table(n, dpf._zlog))
            assert torch.equal(got[perm], want_shares), "expand check failed"
        elif strategy == "coop":
            got = launch(1, keys_gpu[:1])[0].cpu()
            padded = torch.nn.functional.pad(table, (0, 16 - entry_size))
            want = torch.from_numpy(
                _core.eval_fused_cpu(k1.numpy(), padded.numpy(), prf))
            assert torch.equal(got, want), "coop check failed"
        else:  # naive/bfs write natural order directly
            got = launch(1, keys_gpu[:1])[0].cpu()
            assert torch.equal(got, want_shares), "%s check failed" % strategy

    # clock-ramp warmup
    tw = time.time()
    while time.time() - tw < 0.4:
        launch(batch, keys_gpu)
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        launch(batch, keys_gpu)
    torch.cuda.synchronize()
    throughput = batch * reps / (time.time() - t0)

    torch.cuda.synchronize()
    t0 = time.time()
    lat_reps = 20
    for _ in range(lat_reps):
        launch(1, keys_gpu[:1])
        torch.cuda.synchronize()
    latency_ms = (time.time() - t0) / lat_reps * 1e3

    result = {
        "strategy": strategy,
        "prf": prf_name,
        "num_entries": n,
        "batch_size": batch,
        "entry_size": entry_size,
        "reps": reps,
        "throughput_dpfs_per_sec": round(throughput, 1),
        "latency_ms": round(latency_ms, 4),
        "key_size_bytes": 2096,
        "device": torch.cuda.get_device_name(0),
    }
    print(result)
    return result


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--strategy", default="fused",
                    choices=["fused", "expand", "naive", "bfs", "coop"])
    ap.add_argument("--prf", default="AES128", choices=list(PRF_IDS))
    ap.add_argument("--n", type=int, default=65536)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--entry-size", type=int, default=16)
    ap.add_argument("--reps", type=int, default=10)
    ap.add_argument("--check", action="store_true")
    a = ap.parse_args()
    run(a.strategy, a.prf, a.n, a.batch, a.entry_size, a.reps, a.check)
