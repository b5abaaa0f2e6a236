// Streaming mod-2^32 GEMM for the wide-entry / huge-table PIR path:
//
//     C[j][col] += sum_k A[j][k] * B[k][col]   (mod 2^32)
//
// A = one-hot DPF shares [batch, K] u32, B = the permuted table [K, N] u32,
// C = [batch, N] u32 (zeroed by the caller).
//
// WHY THIS KERNEL EXISTS next to the MFMA GEMM (gemm_u32.hip): the MFMA
// path materializes a transposed copy of B plus 4 int8 digit planes of
// BOTH operands — 2x the table's bytes in extra HBM.  Fine at 16 GB;
// impossible for the 200-288 GB tables MI355X's HBM3E is for.  This
// kernel reads the u32 table IN PLACE, once per <=16-key chunk, at
// streaming bandwidth.
//
// v2 design notes (the v1 of this kernel measured 305 GB/s — 20x off
// the HBM bound — for two reasons, both fixed here):
//   * v1 was INSTRUCTION-bound: one 4-byte load per thread carried
//     `batch` v_mac + `batch` LDS-broadcast reads (~1 cycle/byte at
//     wave64).  v2 gives each lane FOUR columns via one dwordx16-shaped
//     uint4 load, amortizing the LDS reads 4x, and caps the in-register
//     batch at 16 (64 acc VGPRs): ~0.3 cycle/byte — above the 8 TB/s
//     HBM line.  Larger batches re-stream the table per 16-key chunk
//     (4 bandwidth-bound passes beat 1 instruction-bound pass ~6x; truly
//     compute-bound shapes belong to the MFMA path, see gpudpf/ops.py).
//   * v1's K-split combined partials with atomicAdd into the tiny
//     [batch, N] output — thousands of colliding RMWs per element.  v2
//     writes per-segment partials (plain coalesced stores) and reduces
//     them with a second kernel, like gemm128's K-split.
//
// Work decomposition (MI355X: 256 CUs / 8 XCDs, wave64):
//   blockIdx.x = K segment (sized so the grid is >= 4096 single-wave
//                workgroups: 16 waves/CU, enough to hide HBM latency and
//                spread each segment's table slab across the XCD L2s),
//   blockIdx.y = 256-column tile,
//   thread t (of 64) = columns [tile*256 + 4t, +4): one uint4 per k row.
//
// Capability parity: serves the reference's "entry_size > 16" TODO
// (dpf.py:16-24) and its dual-stream expand||matmul benchmark pattern
// (dpf_benchmark.cu:191-231) at table sizes the reference cannot hold.

#include <hip/hip_runtime.h>

#include <mutex>
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

constexpr int kThreads = 64;       // one wave per workgroup
constexpr int kColsPerTile = 256;  // 64 lanes x 4 columns
constexpr int kChunkK = 64;        // staged share rows per LDS refill
constexpr int kMaxB = 16;          // accumulators = 4*B VGPRs per lane

// Grow-only per-device partials scratch (kept alive forever so hipGraphs
// capturing a launch stay valid — same policy as dpf_kernels.hip).
// Shared across launches of one device: concurrent stream-GEMM launches
// on DIFFERENT streams would race on it — the python API serializes
// launches per call, which is the supported pattern (as with the DFS
// scratch in dpf_kernels.hip).
std::mutex g_part_mu;
void* g_part[64] = {};
size_t g_part_bytes[64] = {};

u32* get_partials(size_t bytes) {
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  std::lock_guard<std::mutex> lock(g_part_mu);
  if (g_part_bytes[dev] < bytes) {
    size_t want = g_part_bytes[dev] ? g_part_bytes[dev] : bytes;
    while (want < bytes) want *= 2;
    void* p = nullptr;
    HIP_CHECK(hipMalloc(&p, want));
    g_part[dev] = p;  // old buffer intentionally retired (graph safety)
    g_part_bytes[dev] = want;
  }
  return reinterpret_cast<u32*>(g_part[dev]);
}

template <int B>
__global__ __launch_bounds__(kThreads) void gemm_u32_stream_kernel(
    const u32* __restrict__ a, const u32* __restrict__ b,
    u32* __restrict__ part, int batch, long long K, long long N,
    long long k_seg, long long col_tiles) {
  // One wave64 per workgroup: LDS writes and reads of the same wave are
  // in program order, so the A-chunk staging needs NO barriers, and the
  // NEXT chunk's staging loads issue (into registers) before the current
  // chunk's MAC loop — global A latency hides under the table stream.
  constexpr int KQ = kChunkK / 4;                       // uint4s per row
  constexpr int STG = (B * KQ + kThreads - 1) / kThreads;  // loads/lane
  __shared__ u32 s_a[B][kChunkK];
  const int t = (int)threadIdx.x;
  const long long col = (long long)blockIdx.y * kColsPerTile + 4 * t;
  const long long k0 = (long long)blockIdx.x * k_seg;
  const long long k1 = (k0 + k_seg < K) ? k0 + k_seg : K;

  u32 ax[B], ay[B], az[B], aw[B];
#pragma unroll
  for (int j = 0; j < B; ++j) ax[j] = ay[j] = az[j] = aw[j] = 0;

  // fast paths need 16-byte alignment (multiple-of-4 row widths; ep is
  // always a multiple of 16 in production)
  const bool in_n = ((N & 3) == 0) && (col + 3 < N);
  const bool a_aligned = (K & 3) == 0;

  auto load_chunk = [&](long long kc, uint4* regs) {
#pragma unroll
    for (int s = 0; s < STG; ++s) {
      const int i = t + s * kThreads;
      uint4 v = make_uint4(0u, 0u, 0u, 0u);
      if (i < B * KQ) {
        const int j = i / KQ, kq = i % KQ;
        const long long kk0 = kc + 4 * kq;
        if (j < batch && kk0 < k1) {
          const u32* arow = a + (u64)j * (u64)K + (u64)kk0;
          if (a_aligned && kk0 + 3 < k1) {
            v = *reinterpret_cast<const uint4*>(arow);
          } else {
            v.x = arow[0];
            if (kk0 + 1 < k1) v.y = arow[1];
            if (kk0 + 2 < k1) v.z = arow[2];
            if (kk0 + 3 < k1) v.w = arow[3];
          }
        }
      }
      regs[s] = v;
    }
  };
  auto commit_chunk = [&](const uint4* regs) {
#pragma unroll
    for (int s = 0; s < STG; ++s) {
      const int i = t + s * kThreads;
      if (i < B * KQ)
        reinterpret_cast<uint4*>(&s_a[0][0])[i] = regs[s];
    }
  };

  uint4 regs[STG];
  load_chunk(k0, regs);
  commit_chunk(regs);

  for (long long kc = k0; kc < k1; kc += kChunkK) {
    const int kn = (int)((kc + kChunkK <= k1) ? kChunkK : (k1 - kc));
    // issue next chunk's A loads now; they retire during the MAC loop
    const bool have_next = kc + kChunkK < k1;
    if (have_next) load_chunk(kc + kChunkK, regs);
    if (in_n) {
      const u32* brow = b + (u64)kc * (u64)N + (u64)col;
      if (kn == kChunkK) {
#pragma unroll 8
        for (int kk = 0; kk < kChunkK; ++kk) {
          const uint4 v = *reinterpret_cast<const uint4*>(brow + (u64)kk * N);
#pragma unroll
          for (int j = 0; j < B; ++j) {
            const u32 s = s_a[j][kk];
            ax[j] += s * v.x;
            ay[j] += s * v.y;
            az[j] += s * v.z;
            aw[j] += s * v.w;
          }
        }
      } else {
        for (int kk = 0; kk < kn; ++kk) {
          const uint4 v = *reinterpret_cast<const uint4*>(brow + (u64)kk * N);
#pragma unroll
          for (int j = 0; j < B; ++j) {
            const u32 s = s_a[j][kk];
            ax[j] += s * v.x;
            ay[j] += s * v.y;
            az[j] += s * v.z;
            aw[j] += s * v.w;
          }
        }
      }
    } else if (col < N) {  // ragged tail columns, scalar loads
      const u32* brow = b + (u64)kc * (u64)N;
      for (int kk = 0; kk < kn; ++kk) {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          if (col + q < N) {
            const u32 v = brow[(u64)kk * N + (u64)(col + q)];
#pragma unroll
            for (int j = 0; j < B; ++j) {
              const u32 s = s_a[j][kk] * v;
              if (q == 0) ax[j] += s;
              if (q == 1) ay[j] += s;
              if (q == 2) az[j] += s;
              if (q == 3) aw[j] += s;
            }
          }
        }
      }
    }
    if (have_next) commit_chunk(regs);
  }
  // per-segment partials: part[seg][j][N-tilewise] — plain stores
  u32* prow = part + ((u64)blockIdx.x * (u64)batch) * (u64)N;
#pragma unroll
  for (int j = 0; j < B; ++j) {
    if (j >= batch) break;
    u32* dst = prow + (u64)j * (u64)N + (u64)col;
    if (in_n) {
      dst[0] = ax[j]; dst[1] = ay[j]; dst[2] = az[j]; dst[3] = aw[j];
    } else if (col < N) {
      dst[0] = ax[j];
      if (col + 1 < N) dst[1] = ay[j];
      if (col + 2 < N) dst[2] = az[j];
      if (col + 3 < N) dst[3] = aw[j];
    }
  }
}

__global__ __launch_bounds__(256) void gemm_u32_stream_reduce_kernel(
    const u32* __restrict__ part, u32* __restrict__ c, long long elems,
    long long segs) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= elems) return;
  u32 s = 0;
  for (long long g = 0; g < segs; ++g) s += part[(u64)g * (u64)elems + (u64)i];
  c[i] += s;  // accumulate into caller's (zeroed or chunk-owned) output
}

template <int B>
void launch_b(const u32* a, const u32* b, u32* c, int batch, long long K,
              long long N, hipStream_t st) {
  const long long col_tiles = (N + kColsPerTile - 1) / kColsPerTile;
  // k-split: >= 4096 single-wave workgroups (16 waves/CU)
  long long segs = (4096 + col_tiles - 1) / col_tiles;
  long long max_segs = (K + kChunkK - 1) / kChunkK;
  if (segs > max_segs) segs = max_segs;
  // bound the partials scratch at 1 GiB
  const long long max_by_mem = ((long long)1 << 30) / ((long long)batch * N * 4);
  if (segs > max_by_mem && max_by_mem >= 1) segs = max_by_mem;
  if (segs < 1) segs = 1;
  long long k_seg = ((K + segs - 1) / segs + kChunkK - 1) / kChunkK * kChunkK;
  segs = (K + k_seg - 1) / k_seg;
  u32* part = get_partials((size_t)segs * batch * N * 4);
  hipLaunchKernelGGL(gemm_u32_stream_kernel<B>,
                     dim3((unsigned)segs, (unsigned)col_tiles),
                     dim3(kThreads), 0, st, a, b, part, batch, K, N, k_seg,
                     col_tiles);
  HIP_CHECK(hipGetLastError());
  const long long elems = (long long)batch * N;
  hipLaunchKernelGGL(gemm_u32_stream_reduce_kernel,
                     dim3((unsigned)((elems + 255) / 256)), dim3(256), 0, st,
                     part, c, elems, segs);
  HIP_CHECK(hipGetLastError());
}

}  // namespace

void launch_gemm_u32_stream(std::uintptr_t a, std::uintptr_t b,
                            std::uintptr_t c, int batch, long long K,
                            long long N, std::uintptr_t stream) {
  if (batch < 1 || batch > kMaxB)
    throw std::invalid_argument("stream GEMM batch must be 1..16 "
                                "(python wrapper chunks larger batches)");
  auto* ap = reinterpret_cast<const u32*>(a);
  auto* bp = reinterpret_cast<const u32*>(b);
  auto* cp = reinterpret_cast<u32*>(c);
  auto st = reinterpret_cast<hipStream_t>(stream);
  if (batch <= 4) launch_b<4>(ap, bp, cp, batch, K, N, st);
  else if (batch <= 8) launch_b<8>(ap, bp, cp, batch, K, N, st);
  else launch_b<16>(ap, bp, cp, batch, K, N, st);
}

}  // namespace gpudpf_hip
