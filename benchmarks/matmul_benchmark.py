"""GEMM128 benchmark (the reference's matmul_benchmark.cu analog):
throughput of the exact mod-2^128 GEMM at PIR-shaped sizes
(M=batch, N=entry words, K=table entries), dict-line output."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import time

import torch

from gpudpf import ops


def bench_u32_mfma(a):
    import numpy as np

    torch.manual_seed(0)
    A = torch.randint(-(2**31), 2**31 - 1, (a.m, a.k), dtype=torch.int64).to(
        torch.int32
    )
    B = torch.randint(-(2**31), 2**31 - 1, (a.k, a.n), dtype=torch.int64).to(
        torch.int32
    )
    if a.check:
        got = ops.pir_matmul_u32(A, B).cpu().numpy()
        want = (A.numpy().astype(np.int64) @ B.numpy().astype(np.int64)).astype(
            np.uint32
        ).astype(np.int32)
        assert np.array_equal(got, want), "u32 mfma check failed"
        print("check OK")
    dev = torch.device("cuda:0")
    A_g, B_g = A.to(dev), B.to(dev)
    ops.pir_matmul_u32(A_g, B_g)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(a.reps):
        ops.pir_matmul_u32(A_g, B_g)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / a.reps
    macs = a.m * a.n * a.k
    print({
        "kernel": "gemm_u32_mfma",
        "m": a.m, "n": a.n, "k": a.k, "reps": a.reps,
        "time_ms": round(dt * 1e3, 3),
        "gmacs32_per_sec": round(macs / dt / 1e9, 2),
    })


def bench_u32_stream(a):
    import numpy as np

    torch.manual_seed(0)
    A = torch.randint(-(2**31), 2**31 - 1, (a.m, a.k), dtype=torch.int64).to(
        torch.int32)
    B = torch.randint(-(2**31), 2**31 - 1, (a.k, a.n), dtype=torch.int64).to(
        torch.int32)
    if a.check:
        got = ops.pir_matmul_u32_stream(A, B).cpu().numpy()
        want = (A.numpy().astype(np.int64) @ B.numpy().astype(np.int64)
                ).astype(np.uint32).astype(np.int32)
        assert np.array_equal(got, want), "u32 stream check failed"
        print("check OK")
    dev = torch.device("cuda:0")
    A_g, B_g = A.to(dev), B.to(dev)
    out = torch.empty((a.m, a.n), dtype=torch.int32, device=dev)
    ops.pir_matmul_u32_stream(A_g, B_g, out=out)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(a.reps):
        ops.pir_matmul_u32_stream(A_g, B_g, out=out)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / a.reps
    macs = a.m * a.n * a.k
    passes = -(-a.m // 64)
    print({
        "kernel": "gemm_u32_stream",
        "m": a.m, "n": a.n, "k": a.k, "reps": a.reps,
        "time_ms": round(dt * 1e3, 3),
        "gmacs32_per_sec": round(macs / dt / 1e9, 2),
        "table_stream_gbps": round(a.k * a.n * 4 * passes / dt / 1e9, 1),
    })


def bench_library_fp32(a):
    """The reference's ONLY vendor-library GEMM call site is a cuBLAS
    fp32 GemmEx over the float-cast table (dpf_google/benchmark.cu:
    134-154) — exact only while entries < 2^24.  MI355X analog: torch
    fp32 matmul (rocBLAS under ROCm), measured here as the library
    baseline next to our exact mod-2^32 kernels."""
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    A = torch.randn(a.m, a.k, device=dev)
    B = torch.randn(a.k, a.n, device=dev)
    torch.matmul(A, B)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(a.reps):
        torch.matmul(A, B)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / a.reps
    macs = a.m * a.n * a.k
    print({
        "kernel": "rocblas_fp32 (library baseline; NOT exact mod 2^32 — "
                  "lossy above 2^24 like the reference's cublasGemmEx)",
        "m": a.m, "n": a.n, "k": a.k, "reps": a.reps,
        "time_ms": round(dt * 1e3, 3),
        "gmacs_per_sec": round(macs / dt / 1e9, 2),
    })


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--m", type=int, default=512)
    ap.add_argument("--n", type=int, default=16)
    ap.add_argument("--k", type=int, default=65536)
    ap.add_argument("--reps", type=int, default=5)
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--kernel", default="u128",
                    choices=["u128", "u32mfma", "u32stream", "rocblas"])
    a = ap.parse_args()
    if a.kernel == "u32mfma":
        bench_u32_mfma(a)
        return
    if a.kernel == "u32stream":
        bench_u32_stream(a)
        return
    if a.kernel == "rocblas":
        bench_library_fp32(a)
        return

    torch.manual_seed(0)
    A = torch.randint(-(2**31), 2**31 - 1, (a.m, a.k, 4), dtype=torch.int64).to(
        torch.int32
    )
    Bt = torch.randint(-(2**31), 2**31 - 1, (a.n, a.k, 4), dtype=torch.int64).to(
        torch.int32
    )
    if a.check:
        got = ops.gemm128(A, Bt).cpu()
        want = ops.gemm128_cpu(A, Bt)
        assert torch.equal(got, want), "gemm128 check failed"
        print("check OK")

    dev = torch.device("cuda:0")
    A_g, B_g = A.to(dev), Bt.to(dev)
    ops.gemm128(A_g, B_g)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(a.reps):
        ops.gemm128(A_g, B_g)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / a.reps
    macs = a.m * a.n * a.k
    print({
        "kernel": "gemm128",
        "m": a.m, "n": a.n, "k": a.k, "reps": a.reps,
        "time_ms": round(dt * 1e3, 3),
        "gmacs128_per_sec": round(macs / dt / 1e9, 2),
    })


if __name__ == "__main__":
    main()
