// Host-side API of the gpudpf HIP kernel library (implemented in
// dpf_kernels.hip, bound to python in hip_bindings.cc).
#pragma once
#include <cstdint>

namespace gpudpf_hip {

// Fused DPF expansion + table inner product (the production PIR path).
//   keys:  device ptr, int32 [batch][524]   (wire-format keys)
//   table: device ptr, u32 [n][16]          (leaf_perm-reordered rows)
//   out:   device ptr, u32 [batch][16]
//   aes_tabs: device ptr to 5*256 u32 AES tables (only read for AES128)
void launch_fused(std::uintptr_t keys, std::uintptr_t table, std::uintptr_t out,
                  std::uintptr_t aes_tabs, int batch, long long n, int depth,
                  int zlog, int prf, std::uintptr_t stream);

// Full expansion to low-32 one-hot shares, permuted row order
//   out: device ptr, u32 [batch][n]  (row r = leaf_perm(idx))
void launch_expand(std::uintptr_t keys, std::uintptr_t out,
                   std::uintptr_t aes_tabs, int batch, long long n, int depth,
                   int zlog, int prf, std::uintptr_t stream);

// Level-synchronized breadth-first expansion (the reference's
// dpf_breadth_first.cu strategy): depth kernel launches, ping-pong u128
// frontiers in scratch, NATURAL-order low-32 one-hot output
//   out: device ptr, u32 [batch][n]
void launch_bfs(std::uintptr_t keys, std::uintptr_t out,
                std::uintptr_t aes_tabs, int batch, long long n, int depth,
                int prf, std::uintptr_t stream);

// Grid-wide cooperative single-key strategy (the reference's dpf_coop.cu):
// one cooperative launch walks one key's whole tree with a grid sync per
// level.  fused=true MACs the leaves against the permuted table into
// out[16] (zeroed by caller); fused=false writes natural-order one-hot
// low-32 shares to out[n].
void launch_coop(std::uintptr_t keys, std::uintptr_t table, std::uintptr_t out,
                 std::uintptr_t aes_tabs, long long n, int depth, int zlog,
                 int prf, bool fused, std::uintptr_t stream);

// Naive per-leaf oracle (O(n log n) PRFs), natural order output.  Test-only.
void launch_naive(std::uintptr_t keys, std::uintptr_t out,
                  std::uintptr_t aes_tabs, int batch, long long n, int depth,
                  int prf, std::uintptr_t stream);

// PRF speed-of-light microbenchmark: blocks x 256 threads each run a
// dependent chain of `iters` pair expansions; out: u32[blocks*256].
void launch_prf_sol(std::uintptr_t aes_tabs, std::uintptr_t out, int blocks,
                    int iters, int prf, std::uintptr_t stream);

// Unit-test probes (tests/test_gpu_alu.py; analog of the reference's
// dpf_gpu/tests/test_128_bit.cu).  All pointers are device int32/uint32
// buffers; u128 values are 4xu32 limbs little-endian.
void launch_probe_alu(std::uintptr_t a, std::uintptr_t b,
                      std::uintptr_t add_out, std::uintptr_t mul_out,
                      int count, std::uintptr_t stream);
void launch_probe_prf(std::uintptr_t seeds, std::uintptr_t aes_tabs,
                      std::uintptr_t pair0, std::uintptr_t pair1,
                      std::uintptr_t single0, std::uintptr_t single1,
                      std::uintptr_t low0, std::uintptr_t low1, int count,
                      int prf, std::uintptr_t stream);

// Exact u128 GEMM (research harness; gemm128.hip).  a: [M,K] u128 (as
// 4xint32 limbs), bt: [N,K] u128, c: [M,N] u128, partials: device scratch
// of gemm128_ksplit(M,N,K)*M*N u128.
int gemm128_ksplit(long long M, long long N, long long K);
void launch_gemm128(std::uintptr_t a, std::uintptr_t bt, std::uintptr_t c,
                    std::uintptr_t partials, long long M, long long N,
                    long long K, std::uintptr_t stream);

// MFMA mod-2^32 GEMM (gemm_u32.hip): digit decomposition + i8 matrix
// cores.  da/dbt: [4][M][K] / [4][N][K] int8 digit planes; c: [M][N] u32
// (zeroed by caller; K-split partials combine with wrapping atomicAdd).
void launch_digits(std::uintptr_t in, std::uintptr_t out, long long count,
                   std::uintptr_t stream);
void launch_gemm_u32_mfma(std::uintptr_t da, std::uintptr_t dbt,
                          std::uintptr_t c, long long M, long long N,
                          long long K, std::uintptr_t stream);

// Streaming mod-2^32 GEMM (gemm_stream.hip): reads the u32 table in place
// (no transposed copy, no digit planes) — the huge-table wide-entry path.
// a: [batch, K] u32 shares; b: [K, N] u32 table; c: [batch, N] u32
// (zeroed by caller).  batch <= 64 (python wrapper chunks larger).
void launch_gemm_u32_stream(std::uintptr_t a, std::uintptr_t b,
                            std::uintptr_t c, int batch, long long K,
                            long long N, std::uintptr_t stream);

}  // namespace gpudpf_hip
