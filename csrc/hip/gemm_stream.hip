// Streaming mod-2^32 GEMM for the wide-entry / huge-table PIR path:
//
//     C[j][col] += sum_k A[j][k] * B[k][col]   (mod 2^32)
//
// A = one-hot DPF shares [batch, K] u32, B = the permuted table [K, N] u32,
// C = [batch, N] u32 (zeroed by the caller; K-split partials combine with
// wrapping atomicAdd).
//
// WHY THIS KERNEL EXISTS next to the MFMA GEMM (gemm_u32.hip): the MFMA
// path materializes a transposed copy of B plus 4 int8 digit planes of
// BOTH operands — 2x the table's bytes in extra HBM.  Fine at 16 GB;
// impossible for the 200-288 GB tables MI355X's HBM3E is for.  This
// kernel reads the u32 table IN PLACE, once per batch, at streaming
// bandwidth.  In the huge-table regime the job is memory-bound anyway:
// with batch <= ~64 the arithmetic is `batch` v_mad_u32 per 4-byte table
// word — far below the VALU ceiling while the 8 TB/s table stream is
// saturated, so matrix cores would not make it faster (they would only
// add the digit-plane traffic).  Batches beyond 64 are chunked by the
// python wrapper (each chunk re-streams the table; crossover to the MFMA
// path for small tables is handled in gpudpf/ops.py).
//
// Work decomposition (MI355X: 256 CUs / 8 XCDs, wave64):
//   blockIdx.x = K segment (k-split sized so the grid is >= 2048
//                workgroups: fills all CUs and spreads every K segment's
//                table slab across the 8 XCD L2s),
//   blockIdx.y = 256-column tile,
//   thread t   = one output column; holds all `batch` accumulators in
//                VGPRs (B_MAX <= 64 keeps the register budget at 2+
//                waves/SIMD — plenty for a bandwidth-bound loop).
// Each k-chunk stages A[0..b)[k-chunk] through LDS once per workgroup;
// the inner loop reads one coalesced 1 KiB row slab of B per k and does
// B uniform-broadcast LDS reads + B v_mad_u32 per thread.
//
// Capability parity: this serves the reference's "entry_size > 16"
// TODO (dpf.py:16-24) and its dual-stream expand||matmul benchmark
// (dpf_benchmark.cu:191-231) at table sizes the reference cannot hold.

#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "dpf_hip_api.h"

namespace gpudpf_hip {

using u32 = std::uint32_t;
using u64 = std::uint64_t;

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess)                                                 \
      throw std::runtime_error(std::string("HIP error: ") +              \
                               hipGetErrorString(_e) + " at " __FILE__   \
                               ":" + std::to_string(__LINE__));          \
  } while (0)

namespace {

constexpr int kColsPerTile = 256;  // == workgroup size
constexpr int kChunkK = 32;        // staged share rows per LDS refill

template <int B>
__global__ __launch_bounds__(kColsPerTile) void gemm_u32_stream_kernel(
    const u32* __restrict__ a, const u32* __restrict__ b, u32* __restrict__ c,
    int batch, long long K, long long N, long long k_seg) {
  __shared__ u32 s_a[B][kChunkK];
  const int t = (int)threadIdx.x;
  const long long col = (long long)blockIdx.y * kColsPerTile + t;
  const long long k0 = (long long)blockIdx.x * k_seg;
  const long long k1 = (k0 + k_seg < K) ? k0 + k_seg : K;
  if (k0 >= K) return;

  u32 acc[B];
#pragma unroll
  for (int j = 0; j < B; ++j) acc[j] = 0;

  for (long long kc = k0; kc < k1; kc += kChunkK) {
    const int kn = (int)((kc + kChunkK <= k1) ? kChunkK : (k1 - kc));
    __syncthreads();
    // cooperative stage of A[:, kc:kc+kn] (rows beyond `batch` are 0)
    for (int i = t; i < B * kChunkK; i += kColsPerTile) {
      const int j = i / kChunkK, kk = i % kChunkK;
      s_a[j][kk] = (j < batch && kk < kn)
                       ? a[(u64)j * (u64)K + (u64)(kc + kk)]
                       : 0u;
    }
    __syncthreads();
    if (col < N) {
      const u32* brow = b + (u64)kc * (u64)N + (u64)col;
      for (int kk = 0; kk < kn; ++kk) {
        const u32 v = brow[(u64)kk * (u64)N];
#pragma unroll
        for (int j = 0; j < B; ++j) acc[j] += s_a[j][kk] * v;
      }
    }
  }
  if (col < N) {
#pragma unroll
    for (int j = 0; j < B; ++j)
      if (j < batch) atomicAdd(c + (u64)j * (u64)N + (u64)col, acc[j]);
  }
}

template <int B>
void launch_b(const u32* a, const u32* b, u32* c, int batch, long long K,
              long long N, hipStream_t st) {
  const long long col_tiles = (N + kColsPerTile - 1) / kColsPerTile;
  // k-split: >= 2048 workgroups to fill 256 CUs 8-deep, segments rounded
  // to the kChunkK stage granularity
  long long segs = (2048 + col_tiles - 1) / col_tiles;
  long long max_segs = (K + kChunkK - 1) / kChunkK;
  if (segs > max_segs) segs = max_segs;
  if (segs < 1) segs = 1;
  long long k_seg = ((K + segs - 1) / segs + kChunkK - 1) / kChunkK * kChunkK;
  segs = (K + k_seg - 1) / k_seg;
  hipLaunchKernelGGL(gemm_u32_stream_kernel<B>,
                     dim3((unsigned)segs, (unsigned)col_tiles),
                     dim3(kColsPerTile), 0, st, a, b, c, batch, K, N, k_seg);
  HIP_CHECK(hipGetLastError());
}

}  // namespace

void launch_gemm_u32_stream(std::uintptr_t a, std::uintptr_t b,
                            std::uintptr_t c, int batch, long long K,
                            long long N, std::uintptr_t stream) {
  if (batch < 1 || batch > 64)
    throw std::invalid_argument("stream GEMM batch must be 1..64");
  auto* ap = reinterpret_cast<const u32*>(a);
  auto* bp = reinterpret_cast<const u32*>(b);
  auto* cp = reinterpret_cast<u32*>(c);
  auto st = reinterpret_cast<hipStream_t>(stream);
  if (batch <= 4) launch_b<4>(ap, bp, cp, batch, K, N, st);
  else if (batch <= 8) launch_b<8>(ap, bp, cp, batch, K, N, st);
  else if (batch <= 16) launch_b<16>(ap, bp, cp, batch, K, N, st);
  else if (batch <= 32) launch_b<32>(ap, bp, cp, batch, K, N, st);
  else launch_b<64>(ap, bp, cp, batch, K, N, st);
}

}  // namespace gpudpf_hip
