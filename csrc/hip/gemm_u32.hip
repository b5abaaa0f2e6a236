// Mod-2^32 integer GEMM on the MFMA matrix cores (gfx950).
//
//   C[M][N] = A[M][K] * B[K][N]  (all u32, product mod 2^32)
//
// This is the PIR "share x table" reduction as a standalone matrix op —
// the role cuBLAS GemmEx plays in the reference's CPU-expansion pipeline
// (paper/kernel/cpu/dpf_google/benchmark.cu:134-154) — mapped onto CDNA4's
// integer matrix cores.  gfx950 MFMA integer inputs are i8 only
// (v_mfma_i32_16x16x64_i8), so each u32 operand is decomposed into four
// SIGNED base-256 digits d_p in [-128,127] (borrow-propagated, exact):
//
//   v = sum_p d_p 256^p  (mod 2^32)  =>
//   C mod 2^32 = sum_{p+q<=3} (sum_k dA_p dB_q) << 8(p+q)   (mod 2^32)
//
// i.e. 10 MFMA accumulator chains per C tile.  i32 MFMA accumulation
// wraps (verified by tests/test_gemm_u32.py with K large enough to
// overflow), and every epilogue shift/add also wraps, so the result is
// exact mod 2^32.  Effective rate: i8 MFMA peak / 10 ~= 5x the v_mad_u32
// VALU path.  K is split over blockIdx.z; partials combine with wrapping
// atomicAdd.

#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "dpf_hip_api.h"

namespace gpudpf_hip {

namespace {

using u32 = std::uint32_t;
using s8 = signed char;
typedef int v4i __attribute__((ext_vector_type(4)));

#define HIP_CHECK_U(expr)                                                 \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess)                                                 \
      throw std::runtime_error(std::string("HIP error: ") +              \
                               hipGetErrorString(_e));                   \
  } while (0)

// u32 -> four signed base-256 digits with borrow propagation.
__global__ void digits_kernel(const u32* __restrict__ in,
                              s8* __restrict__ out, long long count,
                              long long plane_stride) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < count; i += stride) {
    u32 v = in[i];
    int carry = 0;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      int t = (int)((v >> (8 * p)) & 0xff) + carry;
      int s = t & 0xff;
      int d = (s >= 128) ? s - 256 : s;
      carry = (t >> 8) + ((s >= 128) ? 1 : 0);
      out[p * plane_stride + i] = (s8)d;
    }
  }
}

// grid (M/64, N/16, ksplit); 256 threads = 4 waves, wave w owns the
// 16-row sub-tile m0+16w.  Fragment layout for v_mfma_i32_16x16x64_i8:
// lane l holds A[row=l&15][k = (l>>4)*16 .. +15] and B[k same][col=l&15];
// C/D: col=l&15, row=(l>>4)*4+reg.
__global__ __launch_bounds__(256) void gemm_u32_mfma_kernel(
    const s8* __restrict__ dA,   // [4][M][K]
    const s8* __restrict__ dBt,  // [4][N][K]
    u32* __restrict__ C,         // [M][N]
    long long M, long long N, long long K, long long kchunk) {
  const int wave = (int)threadIdx.x >> 6;
  const int lane = (int)threadIdx.x & 63;
  const long long m0 = (long long)blockIdx.x * 64 + wave * 16;
  const long long n0 = (long long)blockIdx.y * 16;
  const int fr = lane & 15;       // A row / B col within the tile
  const int kg = lane >> 4;       // k-group 0..3 (16 bytes each)
  const long long kb = (long long)blockIdx.z * kchunk;
  const long long ke = (kb + kchunk < K) ? kb + kchunk : K;

  v4i acc[10];
#pragma unroll
  for (int i = 0; i < 10; ++i) acc[i] = v4i{0, 0, 0, 0};

  const long long a_row = (m0 + fr) * K;
  const long long b_row = (n0 + fr) * K;
  for (long long k0 = kb; k0 < ke; k0 += 64) {
    v4i af[4], bf[4];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      af[p] = *reinterpret_cast<const v4i*>(dA + p * M * K + a_row + k0 +
                                            kg * 16);
      bf[p] = *reinterpret_cast<const v4i*>(dBt + p * N * K + b_row + k0 +
                                            kg * 16);
    }
    int idx = 0;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
#pragma unroll
      for (int q = 0; q + p < 4; ++q) {
        acc[idx] = __builtin_amdgcn_mfma_i32_16x16x64_i8(af[p], bf[q],
                                                         acc[idx], 0, 0, 0);
        ++idx;
      }
    }
  }

  // shift per accumulator: pairs in (p,q) order
  // p=0: q=0..3 -> s=0,8,16,24 ; p=1: q=0..2 -> 8,16,24 ; p=2: 16,24 ; p=3: 24
  const int shifts[10] = {0, 8, 16, 24, 8, 16, 24, 16, 24, 24};
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    u32 c = 0;
#pragma unroll
    for (int i = 0; i < 10; ++i) c += ((u32)acc[i][r]) << shifts[i];
    const long long row = m0 + (lane >> 4) * 4 + r;
    atomicAdd(&C[row * N + n0 + fr], c);
  }
}

}  // namespace

void launch_digits(std::uintptr_t in, std::uintptr_t out, long long count,
                   std::uintptr_t stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  long long blocks = (count + 255) / 256;
  if (blocks > (1 << 20)) blocks = 1 << 20;  // grid-stride beyond this
  hipLaunchKernelGGL(digits_kernel, dim3((unsigned)blocks), dim3(256), 0, s,
                     reinterpret_cast<const u32*>(in),
                     reinterpret_cast<s8*>(out), count, count);
  HIP_CHECK_U(hipGetLastError());
}

void launch_gemm_u32_mfma(std::uintptr_t da, std::uintptr_t dbt,
                          std::uintptr_t c, long long M, long long N,
                          long long K, std::uintptr_t stream) {
  if (M % 64 || N % 16 || K % 64)
    throw std::invalid_argument("gemm_u32: M%64, N%16, K%64 must be 0");
  long long xy = (M / 64) * (N / 16);
  long long want = (1024 + xy - 1) / xy;
  long long maxs = K / 64;
  long long ks = 1;
  while (ks * 2 <= want && ks * 2 <= maxs) ks *= 2;
  long long kchunk = ((K / ks) + 63) / 64 * 64;
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(gemm_u32_mfma_kernel,
                     dim3((unsigned)(M / 64), (unsigned)(N / 16),
                          (unsigned)ks),
                     dim3(256), 0, s, reinterpret_cast<const s8*>(da),
                     reinterpret_cast<const s8*>(dbt),
                     reinterpret_cast<u32*>(c), M, N, K, kchunk);
  HIP_CHECK_U(hipGetLastError());
}

}  // namespace gpudpf_hip
