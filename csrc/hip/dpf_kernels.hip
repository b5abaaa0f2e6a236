// MI355X (gfx950, CDNA4) HIP kernels for DPF expansion + fused PIR lookup.
//
// Design (MI355X-first, not a port — see SURVEY.md §7):
//   * One key per workgroup of Z = 1<<zlog threads (Z = 256 = 4 wave64 for
//     n >= 512).  Grid = batch, so a 512-key batch puts 2 workgroups on
//     each of the 256 CUs.
//   * Phase 1: breadth-first expansion of the GGM root to Z frontier seeds
//     through an LDS ping-pong (zlog levels, ~2Z PRFs — negligible).
//   * Phase 2: each thread owns ONE subtree and walks it with a sibling-stack
//     DFS kept in LDS (stack index is wave-uniform per step, so ds_read/
//     ds_write are conflict-free b128 ops).  All threads execute the same
//     DFS schedule => zero divergence.  Each visited interior node expands
//     BOTH children from one parent (for AES this shares one key schedule
//     across the two encryptions — the reference re-expands per call,
//     dpf_gpu/prf/prf.cu:159-184, flagged in its own TODO dpf.py:32-33).
//   * Leaves: only the low 32 bits of a leaf share contribute to the
//     (mod 2^32-truncated) output, because truncation is a ring hom of
//     Z_2^128 -> Z_2^32.  The fused MAC is therefore 16 v_mad_u32 per leaf
//     against u32 table rows — 16x less arithmetic and 4x less table
//     traffic than the reference's mod-2^128 MAC (dpf_hybrid.cu:166-172),
//     with bit-identical output.
//   * Table layout: row(idx) = j<<(zlog+1) | t<<1 | b (leaf_perm in
//     csrc/core/dpf_core.cc), so at DFS step j the workgroup reads one
//     contiguous 32 KB slab and every lane reads its two rows as 8
//     contiguous dwordx4 loads.  All workgroups stream the table in the
//     same order => cross-key reuse in L2/LLC.
//
// PRF device implementations mirror csrc/core/prf.cc bit-exactly (tested
// by tests/test_gpu.py against the CPU core).

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

// ---------------------------------------------------------------------------
// 128-bit helpers (uint4 limbs, x = bits 31..0 ... w = bits 127..96)
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint4 add128(uint4 a, uint4 b) {
  u64 alo = ((u64)a.y << 32) | a.x, ahi = ((u64)a.w << 32) | a.z;
  u64 blo = ((u64)b.y << 32) | b.x, bhi = ((u64)b.w << 32) | b.z;
  u64 lo = alo + blo;
  u64 hi = ahi + bhi + (lo < alo ? 1u : 0u);
  return make_uint4((u32)lo, (u32)(lo >> 32), (u32)hi, (u32)(hi >> 32));
}

__device__ __forceinline__ u32 rotl(u32 v, int s) {
  return (v << s) | (v >> (32 - s));  // lowers to v_alignbit_b32
}

// ---------------------------------------------------------------------------
// PRF constants / ids (wire values match the reference: prf.cu:6-9)
// ---------------------------------------------------------------------------
#define PRF_DUMMY 0
#define PRF_SALSA20 1
#define PRF_CHACHA20 2
#define PRF_AES128 3

// ---------------------------------------------------------------------------
// DUMMY: seed*(pos+4242) + (pos+4242) over Z_2^128
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint4 prf_dummy_full(uint4 seed, u32 pos) {
  unsigned __int128 s = ((unsigned __int128)(((u64)seed.w << 32) | seed.z) << 64) |
                        (((u64)seed.y << 32) | seed.x);
  unsigned __int128 m = (unsigned __int128)(pos + 4242u);
  unsigned __int128 r = s * m + m;
  u64 lo = (u64)r, hi = (u64)(r >> 64);
  return make_uint4((u32)lo, (u32)(lo >> 32), (u32)hi, (u32)(hi >> 32));
}

// ---------------------------------------------------------------------------
// Salsa20/12 (conventions: seed words 1..4 high->low, pos word 9, output
// words 1..4 — see csrc/core/prf.cc)
// ---------------------------------------------------------------------------
#define SALSA_QR(a, b, c, d)  \
  b ^= rotl(a + d, 7);        \
  c ^= rotl(b + a, 9);        \
  d ^= rotl(c + b, 13);       \
  a ^= rotl(d + c, 18)

// Runs the 12 rounds; returns the four result words (x[1..4]+in[1..4]).
__device__ __forceinline__ uint4 salsa12_core(uint4 seed, u32 pos) {
  const u32 c0 = 0x65787061u, c5 = 0x6e642033u, c10 = 0x322d6279u,
            c15 = 0x7465206bu;
  u32 x0 = c0, x1 = seed.w, x2 = seed.z, x3 = seed.y, x4 = seed.x, x5 = c5,
      x6 = 0, x7 = 0, x8 = 0, x9 = pos, x10 = c10, x11 = 0, x12 = 0, x13 = 0,
      x14 = 0, x15 = c15;
#pragma unroll
  for (int r = 0; r < 6; ++r) {
    SALSA_QR(x0, x4, x8, x12);
    SALSA_QR(x5, x9, x13, x1);
    SALSA_QR(x10, x14, x2, x6);
    SALSA_QR(x15, x3, x7, x11);
    SALSA_QR(x0, x1, x2, x3);
    SALSA_QR(x5, x6, x7, x4);
    SALSA_QR(x10, x11, x8, x9);
    SALSA_QR(x15, x12, x13, x14);
  }
  // result words high->low: out1, out2, out3, out4
  return make_uint4(x4 + seed.x, x3 + seed.y, x2 + seed.z, x1 + seed.w);
}

// ---------------------------------------------------------------------------
// ChaCha20/12 (seed words 4..7 high->low, pos word 13, output words 4..7)
// ---------------------------------------------------------------------------
#define CHACHA_QR(a, b, c, d) \
  a += b;  d ^= a;  d = rotl(d, 16); \
  c += d;  b ^= c;  b = rotl(b, 12); \
  a += b;  d ^= a;  d = rotl(d, 8);  \
  c += d;  b ^= c;  b = rotl(b, 7)

__device__ __forceinline__ uint4 chacha12_core(uint4 seed, u32 pos) {
  u32 x0 = 0x65787061u, x1 = 0x6e642033u, x2 = 0x322d6279u, x3 = 0x7465206bu;
  u32 x4 = seed.w, x5 = seed.z, x6 = seed.y, x7 = seed.x;
  u32 x8 = 0, x9 = 0, x10 = 0, x11 = 0, x12 = 0, x13 = pos, x14 = 0, x15 = 0;
#pragma unroll
  for (int r = 0; r < 6; ++r) {
    CHACHA_QR(x0, x4, x8, x12);
    CHACHA_QR(x1, x5, x9, x13);
    CHACHA_QR(x2, x6, x10, x14);
    CHACHA_QR(x3, x7, x11, x15);
    CHACHA_QR(x0, x5, x10, x15);
    CHACHA_QR(x1, x6, x11, x12);
    CHACHA_QR(x2, x7, x8, x13);
    CHACHA_QR(x3, x4, x9, x14);
  }
  return make_uint4(x7 + seed.x, x6 + seed.y, x5 + seed.z, x4 + seed.w);
}

// ---------------------------------------------------------------------------
// AES-128 with LDS T-tables.  aes_lds layout: te0|te1|te2|te3|sbox, each
// 256 u32 (tables generated host-side by gpudpf::aes128_tables).
// Key schedule is expanded once per NODE and shared by both children.
// ---------------------------------------------------------------------------
__device__ __forceinline__ void aes_expand_rk(uint4 seed, const u32* sb,
                                              u32 rk[44]) {
  rk[0] = __builtin_bswap32(seed.x);
  rk[1] = __builtin_bswap32(seed.y);
  rk[2] = __builtin_bswap32(seed.z);
  rk[3] = __builtin_bswap32(seed.w);
  u32 rcon = 0x01u;
#pragma unroll
  for (int r = 1; r <= 10; ++r) {
    u32 w = rk[4 * r - 1];
    w = (w << 8) | (w >> 24);
    w = (sb[(w >> 24) & 0xff] << 24) | (sb[(w >> 16) & 0xff] << 16) |
        (sb[(w >> 8) & 0xff] << 8) | sb[w & 0xff];
    w ^= (rcon << 24);
    rcon = (rcon << 1) ^ ((rcon & 0x80u) ? 0x11bu : 0u);
    rcon &= 0xffu;
    rk[4 * r] = rk[4 * r - 4] ^ w;
    rk[4 * r + 1] = rk[4 * r - 3] ^ rk[4 * r];
    rk[4 * r + 2] = rk[4 * r - 2] ^ rk[4 * r + 1];
    rk[4 * r + 3] = rk[4 * r - 1] ^ rk[4 * r + 2];
  }
}

// Encrypt the 1-byte counter block (pos in {0,1}); full 128-bit result.
__device__ __forceinline__ uint4 aes_cipher(const u32 rk[44], const u32* tabs,
                                            u32 pos) {
  const u32* te0 = tabs;
  const u32* te1 = tabs + 256;
  const u32* te2 = tabs + 512;
  const u32* te3 = tabs + 768;
  const u32* sb = tabs + 1024;
  u32 s0 = (pos << 24) ^ rk[0], s1 = rk[1], s2 = rk[2], s3 = rk[3];
#pragma unroll
  for (int r = 1; r < 10; ++r) {
    u32 n0 = te0[s0 >> 24] ^ te1[(s1 >> 16) & 0xff] ^ te2[(s2 >> 8) & 0xff] ^
             te3[s3 & 0xff] ^ rk[4 * r];
    u32 n1 = te0[s1 >> 24] ^ te1[(s2 >> 16) & 0xff] ^ te2[(s3 >> 8) & 0xff] ^
             te3[s0 & 0xff] ^ rk[4 * r + 1];
    u32 n2 = te0[s2 >> 24] ^ te1[(s3 >> 16) & 0xff] ^ te2[(s0 >> 8) & 0xff] ^
             te3[s1 & 0xff] ^ rk[4 * r + 2];
    u32 n3 = te0[s3 >> 24] ^ te1[(s0 >> 16) & 0xff] ^ te2[(s1 >> 8) & 0xff] ^
             te3[s2 & 0xff] ^ rk[4 * r + 3];
    s0 = n0; s1 = n1; s2 = n2; s3 = n3;
  }
  u32 o0 = ((sb[s0 >> 24] << 24) | (sb[(s1 >> 16) & 0xff] << 16) |
            (sb[(s2 >> 8) & 0xff] << 8) | sb[s3 & 0xff]) ^ rk[40];
  u32 o1 = ((sb[s1 >> 24] << 24) | (sb[(s2 >> 16) & 0xff] << 16) |
            (sb[(s3 >> 8) & 0xff] << 8) | sb[s0 & 0xff]) ^ rk[41];
  u32 o2 = ((sb[s2 >> 24] << 24) | (sb[(s3 >> 16) & 0xff] << 16) |
            (sb[(s0 >> 8) & 0xff] << 8) | sb[s1 & 0xff]) ^ rk[42];
  u32 o3 = ((sb[s3 >> 24] << 24) | (sb[(s0 >> 16) & 0xff] << 16) |
            (sb[(s1 >> 8) & 0xff] << 8) | sb[s2 & 0xff]) ^ rk[43];
  // ciphertext bytes little-endian -> u128 words
  return make_uint4(__builtin_bswap32(o0), __builtin_bswap32(o1),
                    __builtin_bswap32(o2), __builtin_bswap32(o3));
}

// Low-32 result only (leaf fast path: result.x = bswap(o0) needs 4 final-
// round sbox lookups instead of 16).
__device__ __forceinline__ u32 aes_cipher_low(const u32 rk[44], const u32* tabs,
                                              u32 pos) {
  const u32* te0 = tabs;
  const u32* te1 = tabs + 256;
  const u32* te2 = tabs + 512;
  const u32* te3 = tabs + 768;
  const u32* sb = tabs + 1024;
  u32 s0 = (pos << 24) ^ rk[0], s1 = rk[1], s2 = rk[2], s3 = rk[3];
#pragma unroll
  for (int r = 1; r < 10; ++r) {
    u32 n0 = te0[s0 >> 24] ^ te1[(s1 >> 16) & 0xff] ^ te2[(s2 >> 8) & 0xff] ^
             te3[s3 & 0xff] ^ rk[4 * r];
    u32 n1 = te0[s1 >> 24] ^ te1[(s2 >> 16) & 0xff] ^ te2[(s3 >> 8) & 0xff] ^
             te3[s0 & 0xff] ^ rk[4 * r + 1];
    u32 n2 = te0[s2 >> 24] ^ te1[(s3 >> 16) & 0xff] ^ te2[(s0 >> 8) & 0xff] ^
             te3[s1 & 0xff] ^ rk[4 * r + 2];
    u32 n3 = te0[s3 >> 24] ^ te1[(s0 >> 16) & 0xff] ^ te2[(s1 >> 8) & 0xff] ^
             te3[s2 & 0xff] ^ rk[4 * r + 3];
    s0 = n0; s1 = n1; s2 = n2; s3 = n3;
  }
  u32 o0 = ((sb[s0 >> 24] << 24) | (sb[(s1 >> 16) & 0xff] << 16) |
            (sb[(s2 >> 8) & 0xff] << 8) | sb[s3 & 0xff]) ^ rk[40];
  return __builtin_bswap32(o0);
}

// ---------------------------------------------------------------------------
// PRF dispatch: full single, full pair (shared AES key schedule), low pair
// ---------------------------------------------------------------------------
template <int PRF>
__device__ __forceinline__ uint4 prf_full(uint4 seed, u32 pos, const u32* aes) {
  if constexpr (PRF == PRF_DUMMY) return prf_dummy_full(seed, pos);
  if constexpr (PRF == PRF_SALSA20) return salsa12_core(seed, pos);
  if constexpr (PRF == PRF_CHACHA20) return chacha12_core(seed, pos);
  if constexpr (PRF == PRF_AES128) {
    u32 rk[44];
    aes_expand_rk(seed, aes + 1024, rk);
    return aes_cipher(rk, aes, pos);
  }
}

template <int PRF>
__device__ __forceinline__ void prf_pair(uint4 seed, const u32* aes, uint4& r0,
                                         uint4& r1) {
  if constexpr (PRF == PRF_AES128) {
    u32 rk[44];
    aes_expand_rk(seed, aes + 1024, rk);
    r0 = aes_cipher(rk, aes, 0);
    r1 = aes_cipher(rk, aes, 1);
  } else {
    r0 = prf_full<PRF>(seed, 0, aes);
    r1 = prf_full<PRF>(seed, 1, aes);
  }
}

template <int PRF>
__device__ __forceinline__ void prf_pair_low(uint4 seed, const u32* aes,
                                             u32& r0, u32& r1) {
  if constexpr (PRF == PRF_DUMMY) {
    r0 = seed.x * 4242u + 4242u;
    r1 = seed.x * 4243u + 4243u;
  } else if constexpr (PRF == PRF_SALSA20) {
    r0 = salsa12_core(seed, 0).x;
    r1 = salsa12_core(seed, 1).x;
  } else if constexpr (PRF == PRF_CHACHA20) {
    r0 = chacha12_core(seed, 0).x;
    r1 = chacha12_core(seed, 1).x;
  } else {
    u32 rk[44];
    aes_expand_rk(seed, aes + 1024, rk);
    r0 = aes_cipher_low(rk, aes, 0);
    r1 = aes_cipher_low(rk, aes, 1);
  }
}

// Expand both children of `seed` at eval-level i (adds correction words).
template <int PRF>
__device__ __forceinline__ void expand_pair(uint4 seed, int i,
                                            const uint4* cw_lds, const u32* aes,
                                            uint4& c0, uint4& c1) {
  uint4 p0, p1;
  prf_pair<PRF>(seed, aes, p0, p1);
  const int sel = (int)(seed.x & 1u);
  c0 = add128(p0, cw_lds[sel * 64 + i * 2 + 0]);
  c1 = add128(p1, cw_lds[sel * 64 + i * 2 + 1]);
}

// Leaf expansion: low-32 values of both children at eval-level 0.
template <int PRF>
__device__ __forceinline__ void expand_leaf_low(uint4 seed, const uint4* cw_lds,
                                                const u32* aes, u32& v0,
                                                u32& v1) {
  u32 p0, p1;
  prf_pair_low<PRF>(seed, aes, p0, p1);
  const int sel = (int)(seed.x & 1u);
  v0 = p0 + cw_lds[sel * 64 + 0].x;
  v1 = p1 + cw_lds[sel * 64 + 1].x;
}

// ---------------------------------------------------------------------------
// Main kernel: phase-1 frontier + per-thread sibling-stack DFS.
// FUSED=true: accumulate the table inner product; else write raw shares.
// ---------------------------------------------------------------------------
template <int PRF, bool FUSED>
__global__ __launch_bounds__(256) void dpf_eval_kernel(const int* __restrict__ keys,
                                const u32* __restrict__ table,
                                u32* __restrict__ out,
                                const u32* __restrict__ aes_tabs, int depth,
                                int zlog, long long n) {
  extern __shared__ u32 smem[];
  const int Z = 1 << zlog;
  const int DS = depth - zlog;  // subtree splits per thread (>= 1)
  const int t = (int)threadIdx.x;
  const long long key_base = (long long)blockIdx.x * 524;

  uint4* cw_lds = reinterpret_cast<uint4*>(smem);  // 128 entries
  uint4* pp = cw_lds + 128;                        // 2*Z ping-pong
  uint4* stack = pp + 2 * Z;                       // Z*(DS-1)
  u32* aes_lds =
      reinterpret_cast<u32*>(stack + (size_t)Z * (DS > 1 ? DS - 1 : 0));
  u32* red = aes_lds + (PRF == PRF_AES128 ? 1280 : 0);

  // Stage codewords (512 ints) and AES tables into LDS.
  for (int idx = t; idx < 128; idx += blockDim.x)
    cw_lds[idx] =
        reinterpret_cast<const uint4*>(keys + key_base + 4)[idx];
  if constexpr (PRF == PRF_AES128) {
    for (int idx = t; idx < 1280; idx += blockDim.x) aes_lds[idx] = aes_tabs[idx];
  }
  if (t == 0) {
    const int* rp = keys + key_base + 516;
    pp[0] = make_uint4((u32)rp[0], (u32)rp[1], (u32)rp[2], (u32)rp[3]);
  }
  __syncthreads();

  // Phase 1: root -> Z frontier seeds (frontier position t has the
  // first-consumed index bit as its MSB: t = bitrev(idx & (Z-1))).
  uint4* a = pp;
  uint4* b = pp + Z;
  for (int l = 1; l <= zlog; ++l) {
    const int i = depth - l;
    if (t < (1 << l)) {
      uint4 parent = a[t >> 1];
      uint4 v = prf_full<PRF>(parent, (u32)(t & 1), aes_lds);
      const int sel = (int)(parent.x & 1u);
      b[t] = add128(v, cw_lds[sel * 64 + i * 2 + (t & 1)]);
    }
    __syncthreads();
    uint4* tmp = a;
    a = b;
    b = tmp;
  }
  uint4 cur = a[t];
  __syncthreads();

  // Phase 2: initial descent to the first leaf-pair parent.
  for (int d = 1; d <= DS - 1; ++d) {
    uint4 c0, c1;
    expand_pair<PRF>(cur, DS - d, cw_lds, aes_lds, c0, c1);
    stack[(d - 1) * Z + t] = c1;
    cur = c0;
  }

  u32 acc[16];
#pragma unroll
  for (int w = 0; w < 16; ++w) acc[w] = 0;

  const long long pairs = 1LL << (DS - 1);
  for (long long j = 0; j < pairs; ++j) {
    u32 v0, v1;
    expand_leaf_low<PRF>(cur, cw_lds, aes_lds, v0, v1);

    const long long row = (j << (zlog + 1)) + ((long long)t << 1);
    if constexpr (FUSED) {
      const uint4* rows = reinterpret_cast<const uint4*>(table + row * 16);
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        uint4 r0 = rows[q];
        uint4 r1 = rows[q + 4];
        acc[4 * q + 0] += v0 * r0.x + v1 * r1.x;
        acc[4 * q + 1] += v0 * r0.y + v1 * r1.y;
        acc[4 * q + 2] += v0 * r0.z + v1 * r1.z;
        acc[4 * q + 3] += v0 * r0.w + v1 * r1.w;
      }
    } else {
      u32* orow = out + (u64)blockIdx.x * (u64)n + (u64)row;
      orow[0] = v0;
      orow[1] = v1;
    }

    if (j + 1 == pairs) break;
    // Pop the deepest pending sibling and descend its bit-0 spine.
    const int c = (int)(__ffsll((unsigned long long)(j + 1)) - 1);
    const int dpop = DS - 1 - c;
    cur = stack[(dpop - 1) * Z + t];
    for (int d = dpop + 1; d <= DS - 1; ++d) {
      uint4 c0, c1;
      expand_pair<PRF>(cur, DS - d, cw_lds, aes_lds, c0, c1);
      stack[(d - 1) * Z + t] = c1;
      cur = c0;
    }
  }

  if constexpr (FUSED) {
    // Wave shfl reduction, then cross-wave sum through LDS.
#pragma unroll
    for (int w = 0; w < 16; ++w) {
      u32 v = acc[w];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
      acc[w] = v;
    }
    const int lane = t & 63, wave = t >> 6, nwaves = (int)blockDim.x >> 6;
    if (lane == 0) {
#pragma unroll
      for (int w = 0; w < 16; ++w) red[wave * 16 + w] = acc[w];
    }
    __syncthreads();
    if (t < 16) {
      u32 s = 0;
      for (int wv = 0; wv < nwaves; ++wv) s += red[wv * 16 + t];
      out[(u64)blockIdx.x * 16 + t] = s;
    }
  }
}

// ---------------------------------------------------------------------------
// Naive oracle kernel: EvaluateFlat per (key, leaf), natural order.
// ---------------------------------------------------------------------------
template <int PRF>
__global__ __launch_bounds__(256) void dpf_naive_kernel(const int* __restrict__ keys,
                                 u32* __restrict__ out,
                                 const u32* __restrict__ aes_tabs, int depth,
                                 long long n) {
  extern __shared__ u32 smem[];
  uint4* cw_lds = reinterpret_cast<uint4*>(smem);
  u32* aes_lds = reinterpret_cast<u32*>(cw_lds + 128);
  const int t = (int)threadIdx.x;
  const long long key_base = (long long)blockIdx.x * 524;
  for (int idx = t; idx < 128; idx += blockDim.x)
    cw_lds[idx] = reinterpret_cast<const uint4*>(keys + key_base + 4)[idx];
  if constexpr (PRF == PRF_AES128) {
    for (int idx = t; idx < 1280; idx += blockDim.x) aes_lds[idx] = aes_tabs[idx];
  }
  __syncthreads();

  const int* rp = keys + key_base + 516;
  const long long leaf = (long long)blockIdx.y * blockDim.x + t;
  if (leaf >= n) return;
  uint4 key = make_uint4((u32)rp[0], (u32)rp[1], (u32)rp[2], (u32)rp[3]);
  long long rem = leaf;
  for (int i = depth - 1; i >= 0; --i) {
    const u32 bit = (u32)(rem & 1);
    uint4 v = prf_full<PRF>(key, bit, aes_lds);
    const int sel = (int)(key.x & 1u);
    key = add128(v, cw_lds[sel * 64 + i * 2 + (int)bit]);
    rem >>= 1;
  }
  out[(u64)blockIdx.x * (u64)n + (u64)leaf] = key.x;
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------
namespace {

size_t fused_shmem_bytes(int Z, int DS, int prf) {
  size_t bytes = 128 * 16;                       // codewords
  bytes += (size_t)2 * Z * 16;                   // phase-1 ping-pong
  bytes += (size_t)Z * (DS > 1 ? DS - 1 : 0) * 16;  // DFS sibling stack
  if (prf == PRF_AES128) bytes += 1280 * 4;      // AES T-tables + sbox
  bytes += (size_t)(Z / 64) * 16 * 4;            // cross-wave reduction
  return bytes;
}

template <int PRF, bool FUSED>
void launch_eval_t(const int* keys, const u32* table, u32* out,
                   const u32* aes_tabs, int batch, long long n, int depth,
                   int zlog, hipStream_t stream) {
  const int Z = 1 << zlog;
  const int DS = depth - zlog;
  const size_t shmem = fused_shmem_bytes(Z, DS, PRF);
  auto kern = dpf_eval_kernel<PRF, FUSED>;
  if (shmem > 65536) {
    HIP_CHECK(hipFuncSetAttribute((const void*)kern,
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  (int)shmem));
  }
  hipLaunchKernelGGL(kern, dim3((unsigned)batch), dim3((unsigned)Z), shmem,
                     stream, keys, table, out, aes_tabs, depth, zlog, n);
  HIP_CHECK(hipGetLastError());
}

template <bool FUSED>
void launch_eval_dispatch(std::uintptr_t keys, std::uintptr_t table,
                          std::uintptr_t out, std::uintptr_t aes_tabs,
                          int batch, long long n, int depth, int zlog, int prf,
                          std::uintptr_t stream) {
  if (batch <= 0) return;
  if (depth < 1 || ((long long)1 << depth) != n)
    throw std::invalid_argument("bad depth/n");
  if (zlog < 6 || zlog >= depth)
    throw std::invalid_argument("zlog must be in [6, depth)");
  auto* k = reinterpret_cast<const int*>(keys);
  auto* tb = reinterpret_cast<const u32*>(table);
  auto* o = reinterpret_cast<u32*>(out);
  auto* a = reinterpret_cast<const u32*>(aes_tabs);
  auto s = reinterpret_cast<hipStream_t>(stream);
  switch (prf) {
    case PRF_DUMMY: launch_eval_t<PRF_DUMMY, FUSED>(k, tb, o, a, batch, n, depth, zlog, s); break;
    case PRF_SALSA20: launch_eval_t<PRF_SALSA20, FUSED>(k, tb, o, a, batch, n, depth, zlog, s); break;
    case PRF_CHACHA20: launch_eval_t<PRF_CHACHA20, FUSED>(k, tb, o, a, batch, n, depth, zlog, s); break;
    case PRF_AES128: launch_eval_t<PRF_AES128, FUSED>(k, tb, o, a, batch, n, depth, zlog, s); break;
    default: throw std::invalid_argument("unknown PRF");
  }
}

}  // namespace

void launch_fused(std::uintptr_t keys, std::uintptr_t table, std::uintptr_t out,
                  std::uintptr_t aes_tabs, int batch, long long n, int depth,
                  int zlog, int prf, std::uintptr_t stream) {
  launch_eval_dispatch<true>(keys, table, out, aes_tabs, batch, n, depth, zlog,
                             prf, stream);
}

void launch_expand(std::uintptr_t keys, std::uintptr_t out,
                   std::uintptr_t aes_tabs, int batch, long long n, int depth,
                   int zlog, int prf, std::uintptr_t stream) {
  launch_eval_dispatch<false>(keys, /*table=*/0, out, aes_tabs, batch, n, depth,
                              zlog, prf, stream);
}

void launch_naive(std::uintptr_t keys, std::uintptr_t out,
                  std::uintptr_t aes_tabs, int batch, long long n, int depth,
                  int prf, std::uintptr_t stream) {
  if (batch <= 0) return;
  const int threads = 256;
  dim3 grid((unsigned)batch, (unsigned)((n + threads - 1) / threads));
  const size_t shmem = 128 * 16 + (prf == PRF_AES128 ? 1280 * 4 : 0);
  auto* k = reinterpret_cast<const int*>(keys);
  auto* o = reinterpret_cast<u32*>(out);
  auto* a = reinterpret_cast<const u32*>(aes_tabs);
  auto s = reinterpret_cast<hipStream_t>(stream);
  switch (prf) {
    case PRF_DUMMY:
      hipLaunchKernelGGL(dpf_naive_kernel<PRF_DUMMY>, grid, dim3(threads), shmem, s, k, o, a, depth, n);
      break;
    case PRF_SALSA20:
      hipLaunchKernelGGL(dpf_naive_kernel<PRF_SALSA20>, grid, dim3(threads), shmem, s, k, o, a, depth, n);
      break;
    case PRF_CHACHA20:
      hipLaunchKernelGGL(dpf_naive_kernel<PRF_CHACHA20>, grid, dim3(threads), shmem, s, k, o, a, depth, n);
      break;
    case PRF_AES128:
      hipLaunchKernelGGL(dpf_naive_kernel<PRF_AES128>, grid, dim3(threads), shmem, s, k, o, a, depth, n);
      break;
    default:
      throw std::invalid_argument("unknown PRF");
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace gpudpf_hip
