// Exact 128-bit integer GEMM over Z_2^128 (the reference's standalone
// GEMM128 research kernel, dpf_gpu/matmul/matmul.cu, re-designed for
// CDNA4): C[M x N] = A[M x K] * B^T[N x K] (B supplied row-major by
// column, i.e. the reference's "B column-major" layout).
//
// MI355X notes:
//   * no inline PTX carry chains — 128-bit ops are expressed through
//     unsigned __int128 / 64-bit limbs and lower to v_add_co/v_addc and
//     v_mad_u64_u32 sequences (llvm picks the carry chain);
//   * K is split across blockIdx.z into partial sums reduced by a second
//     kernel (mod 2^128 addition is associative/commutative);
//   * tiles stage through LDS as 16x16 C-tiles with a 64-deep K step
//     (A tile 16 KiB + B tile 16 KiB).
// The mod-2^128 MAC cannot use MFMA (gfx950 integer matrix cores are
// i8-input only); the MFMA path for the mod-2^32 PIR reduction lives in
// gemm_u32 (see csrc/hip/dpf_kernels.hip fused MAC and SURVEY.md).

#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "dpf_hip_api.h"

namespace gpudpf_hip {

namespace {

using u32 = std::uint32_t;
using u64 = std::uint64_t;
using u128 = unsigned __int128;

#define HIP_CHECK_G(expr)                                                 \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess)                                                 \
      throw std::runtime_error(std::string("HIP error: ") +              \
                               hipGetErrorString(_e));                   \
  } while (0)

struct u128v {
  u64 lo, hi;
};

__device__ __forceinline__ u128v add128v(u128v a, u128v b) {
  u64 lo = a.lo + b.lo;
  return {lo, a.hi + b.hi + (lo < a.lo ? 1u : 0u)};
}

// low 128 bits of a*b
__device__ __forceinline__ u128v mul128v(u128v a, u128v b) {
  u128 p = (u128)a.lo * b.lo;
  u64 hi = (u64)(p >> 64) + a.lo * b.hi + a.hi * b.lo;
  return {(u64)p, hi};
}

#define TK 64  // K-step per LDS stage

// grid: (M/16, N/16, ksplit); block 256 threads = one 16x16 C tile.
// partials: [ksplit][M][N] u128.
__global__ __launch_bounds__(256) void gemm128_kernel(
    const u128v* __restrict__ A,   // [M][K]
    const u128v* __restrict__ Bt,  // [N][K]
    u128v* __restrict__ partials, long long M, long long N, long long K,
    long long kchunk) {
  __shared__ u128v a_lds[16][TK];
  __shared__ u128v b_lds[16][TK];
  const int tm = (int)threadIdx.x / 16;
  const int tn = (int)threadIdx.x % 16;
  const long long m0 = (long long)blockIdx.x * 16;
  const long long n0 = (long long)blockIdx.y * 16;
  const long long k_begin = (long long)blockIdx.z * kchunk;
  const long long k_end = (k_begin + kchunk < K) ? k_begin + kchunk : K;

  u128v acc{0, 0};
  for (long long k0 = k_begin; k0 < k_end; k0 += TK) {
    const int kw = (int)((k_end - k0 < TK) ? (k_end - k0) : TK);
    // cooperative stage: 256 threads load 16 rows x TK of A and B
    for (int idx = (int)threadIdx.x; idx < 16 * kw; idx += 256) {
      const int r = idx / kw, c = idx % kw;
      a_lds[r][c] = (m0 + r < M) ? A[(m0 + r) * K + k0 + c] : u128v{0, 0};
      b_lds[r][c] = (n0 + r < N) ? Bt[(n0 + r) * K + k0 + c] : u128v{0, 0};
    }
    __syncthreads();
    for (int c = 0; c < kw; ++c)
      acc = add128v(acc, mul128v(a_lds[tm][c], b_lds[tn][c]));
    __syncthreads();
  }
  if (m0 + tm < M && n0 + tn < N)
    partials[((long long)blockIdx.z * M + m0 + tm) * N + n0 + tn] = acc;
}

__global__ void gemm128_reduce_kernel(const u128v* __restrict__ partials,
                                      u128v* __restrict__ C, long long MN,
                                      int ksplit) {
  const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  u128v acc{0, 0};
  for (int z = 0; z < ksplit; ++z) acc = add128v(acc, partials[z * MN + i]);
  C[i] = acc;
}

}  // namespace

int gemm128_ksplit(long long M, long long N, long long K) {
  // fill ~4096 blocks; K-chunks are multiples of TK
  long long xy = ((M + 15) / 16) * ((N + 15) / 16);
  long long want = (4096 + xy - 1) / xy;
  long long maxs = (K + TK - 1) / TK;
  long long s = 1;
  while (s * 2 <= want && s * 2 <= maxs) s *= 2;
  return (int)s;
}

void launch_gemm128(std::uintptr_t a, std::uintptr_t bt, std::uintptr_t c,
                    std::uintptr_t partials, long long M, long long N,
                    long long K, std::uintptr_t stream) {
  const int ksplit = gemm128_ksplit(M, N, K);
  long long kchunk = ((K + ksplit - 1) / ksplit + TK - 1) / TK * TK;
  dim3 grid((unsigned)((M + 15) / 16), (unsigned)((N + 15) / 16),
            (unsigned)ksplit);
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(gemm128_kernel, grid, dim3(256), 0, s,
                     reinterpret_cast<const u128v*>(a),
                     reinterpret_cast<const u128v*>(bt),
                     reinterpret_cast<u128v*>(partials), M, N, K, kchunk);
  HIP_CHECK_G(hipGetLastError());
  const long long MN = M * N;
  hipLaunchKernelGGL(gemm128_reduce_kernel,
                     dim3((unsigned)((MN + 255) / 256)), dim3(256), 0, s,
                     reinterpret_cast<const u128v*>(partials),
                     reinterpret_cast<u128v*>(c), MN, ksplit);
  HIP_CHECK_G(hipGetLastError());
}

}  // namespace gpudpf_hip
