// MI355X (gfx950, CDNA4) HIP kernels for DPF expansion + fused PIR lookup.
//
// Design (MI355X-first, not a port — see SURVEY.md §7):
//   * One key per workgroup of Z = 1<<zlog threads (Z = 256 = 4 wave64 for
//     n >= 512).  Grid = batch, so a 512-key batch puts >=2 workgroups on
//     each of the 256 CUs.
//   * Phase 1: breadth-first expansion of the GGM root to Z frontier seeds
//     through an LDS ping-pong (zlog levels, ~2Z PRFs — negligible).
//   * Phase 2: each thread owns ONE subtree and walks it with a sibling-
//     stack DFS.  All threads execute the same DFS schedule => zero
//     divergence, and the stack index is wave-uniform per step so LDS
//     accesses are conflict-free b128 ops.  The two HOT stack levels
//     (popped on 3/4 of all steps) live in registers; the cold remainder
//     shares the phase-1 ping-pong LDS region (they are never live at the
//     same time), keeping LDS small enough for 3-4 workgroups per CU.
//   * Each visited interior node expands BOTH children from one parent;
//     for AES this shares one key schedule across the two encryptions
//     (the reference re-expands per call: dpf_gpu/prf/prf.cu:159-184,
//     flagged in its own TODO dpf.py:32-33).
//   * Leaves: only the low 32 bits of a leaf share contribute to the
//     (mod 2^32-truncated) output, because truncation is a ring hom of
//     Z_2^128 -> Z_2^32.  The fused MAC is therefore 16 v_mad_u32 per leaf
//     against u32 table rows — 16x less arithmetic and 4x less table
//     traffic than the reference's mod-2^128 MAC (dpf_hybrid.cu:166-172),
//     with bit-identical output.  Table loads are issued BEFORE the leaf
//     PRFs of the same step, so ~1000 cycles of cipher work hides them.
//   * AES T-table: a single 256-entry table replicated 32-way and
//     interleaved (entry*32 + lane%32) so every lane reads its own LDS
//     bank — measured 4.3x bank-conflict amplification with flat tables
//     (SQ_LDS_BANK_CONFLICT 3.5e10 vs SQ_INSTS_LDS 8.1e9, profiles/).
//     te1..te3 are byte rotations of te0 (1 v_alignbit per lookup) and
//     sbox[x] = byte2 of te0[x], so one table serves the whole cipher.
//   * Table layout: row(idx) = j<<(zlog+1) | t<<1 | b (leaf_perm in
//     csrc/core/dpf_core.cc): at DFS step j the workgroup reads one
//     contiguous 32 KB slab; every lane reads its two rows as 8 contiguous
//     dwordx4 loads.  All workgroups stream the table in the same order
//     => cross-key reuse in L2/LLC.
//
// PRF device implementations mirror csrc/core/prf.cc bit-exactly (tested
// by tests/test_gpu.py against the CPU core).

#include <hip/hip_runtime.h>

#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

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
__device__ __forceinline__ u32 rotr8(u32 v) { return (v >> 8) | (v << 24); }
__device__ __forceinline__ u32 rotr16(u32 v) { return (v >> 16) | (v << 16); }
__device__ __forceinline__ u32 rotr24(u32 v) { return (v >> 24) | (v << 8); }

#define PRF_DUMMY 0
#define PRF_SALSA20 1
#define PRF_CHACHA20 2
#define PRF_AES128 3

// AES replicated-table geometry
#define AES_REP 32
// DFS sibling-stack levels kept in LDS (deeper levels: registers;
// shallower: global scratch)
#define MAX_LDS_LEVELS 4
#define AES_LDS_WORDS (256 * AES_REP)  // 32 KiB

// ---------------------------------------------------------------------------
// DUMMY: seed*(pos+4242) + (pos+4242) over Z_2^128
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint4 prf_dummy_full(uint4 seed, u32 pos) {
  unsigned __int128 s =
      ((unsigned __int128)(((u64)seed.w << 32) | seed.z) << 64) |
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
  return make_uint4(x4 + seed.x, x3 + seed.y, x2 + seed.z, x1 + seed.w);
}

// Both children interleaved in one pass: the two blocks (pos=0 / pos=1)
// are independent dependency chains, so interleaving doubles the ILP the
// SIMD can draw on — the DFS runs at ~2 waves/SIMD where a single chain's
// QR latency is not fully hidden.
#define SALSA_QR2(a, b, c, d, a2, b2, c2, d2) \
  b ^= rotl(a + d, 7);   b2 ^= rotl(a2 + d2, 7);   \
  c ^= rotl(b + a, 9);   c2 ^= rotl(b2 + a2, 9);   \
  d ^= rotl(c + b, 13);  d2 ^= rotl(c2 + b2, 13);  \
  a ^= rotl(d + c, 18);  a2 ^= rotl(d2 + c2, 18)

__device__ __forceinline__ void salsa12_pair_core(uint4 seed, uint4& r0,
                                                  uint4& r1) {
  const u32 c0 = 0x65787061u, c5 = 0x6e642033u, c10 = 0x322d6279u,
            c15 = 0x7465206bu;
  u32 x0 = c0, x1 = seed.w, x2 = seed.z, x3 = seed.y, x4 = seed.x, x5 = c5,
      x6 = 0, x7 = 0, x8 = 0, x9 = 0, x10 = c10, x11 = 0, x12 = 0, x13 = 0,
      x14 = 0, x15 = c15;
  u32 y0 = c0, y1 = x1, y2 = x2, y3 = x3, y4 = x4, y5 = c5, y6 = 0, y7 = 0,
      y8 = 0, y9 = 1, y10 = c10, y11 = 0, y12 = 0, y13 = 0, y14 = 0, y15 = c15;
#pragma unroll
  for (int r = 0; r < 6; ++r) {
    SALSA_QR2(x0, x4, x8, x12, y0, y4, y8, y12);
    SALSA_QR2(x5, x9, x13, x1, y5, y9, y13, y1);
    SALSA_QR2(x10, x14, x2, x6, y10, y14, y2, y6);
    SALSA_QR2(x15, x3, x7, x11, y15, y3, y7, y11);
    SALSA_QR2(x0, x1, x2, x3, y0, y1, y2, y3);
    SALSA_QR2(x5, x6, x7, x4, y5, y6, y7, y4);
    SALSA_QR2(x10, x11, x8, x9, y10, y11, y8, y9);
    SALSA_QR2(x15, x12, x13, x14, y15, y12, y13, y14);
  }
  r0 = make_uint4(x4 + seed.x, x3 + seed.y, x2 + seed.z, x1 + seed.w);
  r1 = make_uint4(y4 + seed.x, y3 + seed.y, y2 + seed.z, y1 + seed.w);
}

// Leaf-level variant: only the LOW output word (out[4] = x4 + seed.x) is
// needed, so the final double-round computes just the dataflow cone of
// x4: row-QR(x5,x6,x7,x4) needs post-column x5 (full QR), x6 (b,c,d of
// its QR), x7 (b,c), x4 (b) — 39 ops instead of 96.
#define SALSA_QR2_HEAD1(a, b, c, d, a2, b2, c2, d2) /* b only */ \
  b ^= rotl(a + d, 7);   b2 ^= rotl(a2 + d2, 7)
#define SALSA_QR2_HEAD2(a, b, c, d, a2, b2, c2, d2) /* b, c */ \
  b ^= rotl(a + d, 7);   b2 ^= rotl(a2 + d2, 7);   \
  c ^= rotl(b + a, 9);   c2 ^= rotl(b2 + a2, 9)
#define SALSA_QR2_HEAD3(a, b, c, d, a2, b2, c2, d2) /* b, c, d */ \
  b ^= rotl(a + d, 7);   b2 ^= rotl(a2 + d2, 7);   \
  c ^= rotl(b + a, 9);   c2 ^= rotl(b2 + a2, 9);   \
  d ^= rotl(c + b, 13);  d2 ^= rotl(c2 + b2, 13)

__device__ __forceinline__ void salsa12_pair_low_core(uint4 seed, u32& lo0,
                                                      u32& lo1) {
  const u32 c0 = 0x65787061u, c5 = 0x6e642033u, c10 = 0x322d6279u,
            c15 = 0x7465206bu;
  u32 x0 = c0, x1 = seed.w, x2 = seed.z, x3 = seed.y, x4 = seed.x, x5 = c5,
      x6 = 0, x7 = 0, x8 = 0, x9 = 0, x10 = c10, x11 = 0, x12 = 0, x13 = 0,
      x14 = 0, x15 = c15;
  u32 y0 = c0, y1 = x1, y2 = x2, y3 = x3, y4 = x4, y5 = c5, y6 = 0, y7 = 0,
      y8 = 0, y9 = 1, y10 = c10, y11 = 0, y12 = 0, y13 = 0, y14 = 0, y15 = c15;
#pragma unroll
  for (int r = 0; r < 5; ++r) {
    SALSA_QR2(x0, x4, x8, x12, y0, y4, y8, y12);
    SALSA_QR2(x5, x9, x13, x1, y5, y9, y13, y1);
    SALSA_QR2(x10, x14, x2, x6, y10, y14, y2, y6);
    SALSA_QR2(x15, x3, x7, x11, y15, y3, y7, y11);
    SALSA_QR2(x0, x1, x2, x3, y0, y1, y2, y3);
    SALSA_QR2(x5, x6, x7, x4, y5, y6, y7, y4);
    SALSA_QR2(x10, x11, x8, x9, y10, y11, y8, y9);
    SALSA_QR2(x15, x12, x13, x14, y15, y12, y13, y14);
  }
  // 6th double-round, pruned to the cone of x4/y4:
  SALSA_QR2_HEAD1(x0, x4, x8, x12, y0, y4, y8, y12);   // new x4 (b)
  SALSA_QR2(x5, x9, x13, x1, y5, y9, y13, y1);         // new x5 (a: full)
  SALSA_QR2_HEAD3(x10, x14, x2, x6, y10, y14, y2, y6); // new x6 (d)
  SALSA_QR2_HEAD2(x15, x3, x7, x11, y15, y3, y7, y11); // new x7 (c)
  SALSA_QR2_HEAD3(x5, x6, x7, x4, y5, y6, y7, y4);     // row: new x4 (d)
  lo0 = x4 + seed.x;
  lo1 = y4 + seed.x;
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

#define CHACHA_QR2(a, b, c, d, a2, b2, c2, d2)                      \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 16);  d2 = rotl(d2, 16);                              \
  c += d;  c2 += d2;  b ^= c;  b2 ^= c2;                            \
  b = rotl(b, 12);  b2 = rotl(b2, 12);                              \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 8);  d2 = rotl(d2, 8);                                \
  c += d;  c2 += d2;  b ^= c;  b2 ^= c2;                            \
  b = rotl(b, 7);  b2 = rotl(b2, 7)

__device__ __forceinline__ void chacha12_pair_core(uint4 seed, uint4& r0,
                                                   uint4& r1) {
  const u32 k0 = 0x65787061u, k1 = 0x6e642033u, k2 = 0x322d6279u,
            k3 = 0x7465206bu;
  u32 x0 = k0, x1 = k1, x2 = k2, x3 = k3;
  u32 x4 = seed.w, x5 = seed.z, x6 = seed.y, x7 = seed.x;
  u32 x8 = 0, x9 = 0, x10 = 0, x11 = 0, x12 = 0, x13 = 0, x14 = 0, x15 = 0;
  u32 y0 = k0, y1 = k1, y2 = k2, y3 = k3;
  u32 y4 = x4, y5 = x5, y6 = x6, y7 = x7;
  u32 y8 = 0, y9 = 0, y10 = 0, y11 = 0, y12 = 0, y13 = 1, y14 = 0, y15 = 0;
#pragma unroll
  for (int r = 0; r < 6; ++r) {
    CHACHA_QR2(x0, x4, x8, x12, y0, y4, y8, y12);
    CHACHA_QR2(x1, x5, x9, x13, y1, y5, y9, y13);
    CHACHA_QR2(x2, x6, x10, x14, y2, y6, y10, y14);
    CHACHA_QR2(x3, x7, x11, x15, y3, y7, y11, y15);
    CHACHA_QR2(x0, x5, x10, x15, y0, y5, y10, y15);
    CHACHA_QR2(x1, x6, x11, x12, y1, y6, y11, y12);
    CHACHA_QR2(x2, x7, x8, x13, y2, y7, y8, y13);
    CHACHA_QR2(x3, x4, x9, x14, y3, y4, y9, y14);
  }
  r0 = make_uint4(x7 + seed.x, x6 + seed.y, x5 + seed.z, x4 + seed.w);
  r1 = make_uint4(y7 + seed.x, y6 + seed.y, y5 + seed.z, y4 + seed.w);
}

// Leaf-level variant: only out[7] = x7 + seed.x is needed.  Final
// double-round cone: diagonal QR(x2,x7,x8,x13) needs its full chain;
// its inputs need x2 (full col QR), x7 (full: b needs a,c,d), x8 (col
// QR c through the second c+=d), x13 (col QR d through the second
// d-rotate) — 56 ops instead of 96.
#define CHACHA_QR2_C10(a, b, c, d, a2, b2, c2, d2) /* c final (skip last b) */ \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 16);  d2 = rotl(d2, 16);                              \
  c += d;  c2 += d2;  b ^= c;  b2 ^= c2;                            \
  b = rotl(b, 12);  b2 = rotl(b2, 12);                              \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 8);  d2 = rotl(d2, 8);                                \
  c += d;  c2 += d2
#define CHACHA_QR2_D10(a, b, c, d, a2, b2, c2, d2) /* d final (skip last c,b) */ \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 16);  d2 = rotl(d2, 16);                              \
  c += d;  c2 += d2;  b ^= c;  b2 ^= c2;                            \
  b = rotl(b, 12);  b2 = rotl(b2, 12);                              \
  a += b;  a2 += b2;  d ^= a;  d2 ^= a2;                            \
  d = rotl(d, 8);  d2 = rotl(d2, 8)

__device__ __forceinline__ void chacha12_pair_low_core(uint4 seed, u32& lo0,
                                                       u32& lo1) {
  const u32 k0 = 0x65787061u, k1 = 0x6e642033u, k2 = 0x322d6279u,
            k3 = 0x7465206bu;
  u32 x0 = k0, x1 = k1, x2 = k2, x3 = k3;
  u32 x4 = seed.w, x5 = seed.z, x6 = seed.y, x7 = seed.x;
  u32 x8 = 0, x9 = 0, x10 = 0, x11 = 0, x12 = 0, x13 = 0, x14 = 0, x15 = 0;
  u32 y0 = k0, y1 = k1, y2 = k2, y3 = k3;
  u32 y4 = x4, y5 = x5, y6 = x6, y7 = x7;
  u32 y8 = 0, y9 = 0, y10 = 0, y11 = 0, y12 = 0, y13 = 1, y14 = 0, y15 = 0;
#pragma unroll
  for (int r = 0; r < 5; ++r) {
    CHACHA_QR2(x0, x4, x8, x12, y0, y4, y8, y12);
    CHACHA_QR2(x1, x5, x9, x13, y1, y5, y9, y13);
    CHACHA_QR2(x2, x6, x10, x14, y2, y6, y10, y14);
    CHACHA_QR2(x3, x7, x11, x15, y3, y7, y11, y15);
    CHACHA_QR2(x0, x5, x10, x15, y0, y5, y10, y15);
    CHACHA_QR2(x1, x6, x11, x12, y1, y6, y11, y12);
    CHACHA_QR2(x2, x7, x8, x13, y2, y7, y8, y13);
    CHACHA_QR2(x3, x4, x9, x14, y3, y4, y9, y14);
  }
  // 6th double-round, pruned to the cone of x7/y7:
  CHACHA_QR2_C10(x0, x4, x8, x12, y0, y4, y8, y12);   // x8 (c)
  CHACHA_QR2_D10(x1, x5, x9, x13, y1, y5, y9, y13);   // x13 (d)
  CHACHA_QR2(x2, x6, x10, x14, y2, y6, y10, y14);     // x2 (a: full)
  CHACHA_QR2(x3, x7, x11, x15, y3, y7, y11, y15);     // x7 (b: full)
  CHACHA_QR2(x2, x7, x8, x13, y2, y7, y8, y13);       // diagonal: x7 (b)
  lo0 = x7 + seed.x;
  lo1 = y7 + seed.x;
}

// ---------------------------------------------------------------------------
// AES-128, replicated-LDS variant (fused kernel).  aes_lds holds te0
// replicated AES_REP-way, interleaved: word(entry e, copy c) at e*32+c.
// Lane l always uses copy l%32 => bank (addr/4)%32 == l%32: conflict-free.
//   te0[x] bytes (MSB..LSB) = [2s, s, s, 3s];  te1 = ror8(te0),
//   te2 = ror16, te3 = ror24;  sbox[x] = byte2 of te0[x].
// ---------------------------------------------------------------------------
struct AesLds {
  const u32* rep;  // replicated te0
  u32 lane;        // threadIdx.x % 32
  __device__ __forceinline__ u32 te0(u32 x) const { return rep[x * AES_REP + lane]; }
  __device__ __forceinline__ u32 sbox(u32 x) const { return (te0(x) >> 16) & 0xff; }
};

// Both children (pos 0 and pos 1) with an ON-THE-FLY key schedule fused
// into the interleaved cipher rounds: the key-expansion sbox chain (4
// dependent LDS lookups per round) overlaps the 32 independent cipher
// lookups of the same round instead of standing alone as a 40-deep serial
// latency chain (PMC: SQ_WAIT_ANY was 31.5% of wave cycles with the
// standalone schedule).  Also drops the rk[44] register array.
template <bool LOW>
__device__ __forceinline__ void aes_cipher_pair(uint4 seed, const AesLds& T,
                                                uint4& ra, uint4& rb) {
  u32 k0 = __builtin_bswap32(seed.x), k1 = __builtin_bswap32(seed.y),
      k2 = __builtin_bswap32(seed.z), k3 = __builtin_bswap32(seed.w);
  u32 s0 = k0, s1 = k1, s2 = k2, s3 = k3;
  u32 u0 = (1u << 24) ^ k0, u1 = k1, u2 = k2, u3 = k3;
  u32 rcon = 0x01u;
  int r_begin = 1;
#ifndef GPUDPF_AES_NOSHARE
  // Shared-prefix rounds: the two children's plaintexts (pos 0 / pos 1)
  // differ in a single byte, so their states agree except for byte
  // (s0 >> 24) until MixColumns diffuses it — round 1 of child b costs
  // ONE extra T-lookup (all four output words share), and round 2 shares
  // 12 of its 16 lookups (each output word reads exactly one byte of the
  // diverged word 0).  Saves ~27 of ~320 lookups per pair; rounds 3+ are
  // fully diverged and run the common loop below.  A/B toggle:
  // compile with -DGPUDPF_AES_NOSHARE for the straight dual-cipher
  // variant (measured comparison in profiles/PROFILING.md).
  {
    // round 1
    u32 w_ = (k3 << 8) | (k3 >> 24);
    w_ = (T.sbox((w_ >> 24) & 0xff) << 24) | (T.sbox((w_ >> 16) & 0xff) << 16) |
         (T.sbox((w_ >> 8) & 0xff) << 8) | T.sbox(w_ & 0xff);
    w_ ^= (rcon << 24);
    rcon = 0x02u;
    k0 ^= w_; k1 ^= k0; k2 ^= k1; k3 ^= k2;
    const u32 A = T.te0(s0 >> 24);
    u32 n0 = A ^ rotr8(T.te0((s1 >> 16) & 0xff)) ^
             rotr16(T.te0((s2 >> 8) & 0xff)) ^ rotr24(T.te0(s3 & 0xff)) ^ k0;
    u32 n1 = T.te0(s1 >> 24) ^ rotr8(T.te0((s2 >> 16) & 0xff)) ^
             rotr16(T.te0((s3 >> 8) & 0xff)) ^ rotr24(T.te0(s0 & 0xff)) ^ k1;
    u32 n2 = T.te0(s2 >> 24) ^ rotr8(T.te0((s3 >> 16) & 0xff)) ^
             rotr16(T.te0((s0 >> 8) & 0xff)) ^ rotr24(T.te0(s1 & 0xff)) ^ k2;
    u32 n3 = T.te0(s3 >> 24) ^ rotr8(T.te0((s0 >> 16) & 0xff)) ^
             rotr16(T.te0((s1 >> 8) & 0xff)) ^ rotr24(T.te0(s2 & 0xff)) ^ k3;
    const u32 m0 = n0 ^ A ^ T.te0((s0 >> 24) ^ 1u);
    // round 2
    w_ = (k3 << 8) | (k3 >> 24);
    w_ = (T.sbox((w_ >> 24) & 0xff) << 24) | (T.sbox((w_ >> 16) & 0xff) << 16) |
         (T.sbox((w_ >> 8) & 0xff) << 8) | T.sbox(w_ & 0xff);
    w_ ^= (rcon << 24);
    rcon = 0x04u;
    k0 ^= w_; k1 ^= k0; k2 ^= k1; k3 ^= k2;
    const u32 c0 = rotr8(T.te0((n1 >> 16) & 0xff)) ^
                   rotr16(T.te0((n2 >> 8) & 0xff)) ^
                   rotr24(T.te0(n3 & 0xff)) ^ k0;
    const u32 c1 = T.te0(n1 >> 24) ^ rotr8(T.te0((n2 >> 16) & 0xff)) ^
                   rotr16(T.te0((n3 >> 8) & 0xff)) ^ k1;
    const u32 c2 = T.te0(n2 >> 24) ^ rotr8(T.te0((n3 >> 16) & 0xff)) ^
                   rotr24(T.te0(n1 & 0xff)) ^ k2;
    const u32 c3 = T.te0(n3 >> 24) ^ rotr16(T.te0((n1 >> 8) & 0xff)) ^
                   rotr24(T.te0(n2 & 0xff)) ^ k3;
    s0 = T.te0(n0 >> 24) ^ c0;
    u0 = T.te0(m0 >> 24) ^ c0;
    s1 = rotr24(T.te0(n0 & 0xff)) ^ c1;
    u1 = rotr24(T.te0(m0 & 0xff)) ^ c1;
    s2 = rotr16(T.te0((n0 >> 8) & 0xff)) ^ c2;
    u2 = rotr16(T.te0((m0 >> 8) & 0xff)) ^ c2;
    s3 = rotr8(T.te0((n0 >> 16) & 0xff)) ^ c3;
    u3 = rotr8(T.te0((m0 >> 16) & 0xff)) ^ c3;
    r_begin = 3;
  }
#endif
#pragma unroll
  for (int r = r_begin; r < 10; ++r) {
    // next round key (k0..k3 become rk[4r..4r+3])
    u32 w_ = (k3 << 8) | (k3 >> 24);
    w_ = (T.sbox((w_ >> 24) & 0xff) << 24) | (T.sbox((w_ >> 16) & 0xff) << 16) |
         (T.sbox((w_ >> 8) & 0xff) << 8) | T.sbox(w_ & 0xff);
    w_ ^= (rcon << 24);
    rcon = (rcon << 1) ^ ((rcon & 0x80u) ? 0x11bu : 0u);
    rcon &= 0xffu;
    k0 ^= w_; k1 ^= k0; k2 ^= k1; k3 ^= k2;
    // cipher round r for both children
    u32 n0 = T.te0(s0 >> 24) ^ rotr8(T.te0((s1 >> 16) & 0xff)) ^
             rotr16(T.te0((s2 >> 8) & 0xff)) ^ rotr24(T.te0(s3 & 0xff)) ^ k0;
    u32 m0 = T.te0(u0 >> 24) ^ rotr8(T.te0((u1 >> 16) & 0xff)) ^
             rotr16(T.te0((u2 >> 8) & 0xff)) ^ rotr24(T.te0(u3 & 0xff)) ^ k0;
    u32 n1 = T.te0(s1 >> 24) ^ rotr8(T.te0((s2 >> 16) & 0xff)) ^
             rotr16(T.te0((s3 >> 8) & 0xff)) ^ rotr24(T.te0(s0 & 0xff)) ^ k1;
    u32 m1 = T.te0(u1 >> 24) ^ rotr8(T.te0((u2 >> 16) & 0xff)) ^
             rotr16(T.te0((u3 >> 8) & 0xff)) ^ rotr24(T.te0(u0 & 0xff)) ^ k1;
    u32 n2 = T.te0(s2 >> 24) ^ rotr8(T.te0((s3 >> 16) & 0xff)) ^
             rotr16(T.te0((s0 >> 8) & 0xff)) ^ rotr24(T.te0(s1 & 0xff)) ^ k2;
    u32 m2 = T.te0(u2 >> 24) ^ rotr8(T.te0((u3 >> 16) & 0xff)) ^
             rotr16(T.te0((u0 >> 8) & 0xff)) ^ rotr24(T.te0(u1 & 0xff)) ^ k2;
    u32 n3 = T.te0(s3 >> 24) ^ rotr8(T.te0((s0 >> 16) & 0xff)) ^
             rotr16(T.te0((s1 >> 8) & 0xff)) ^ rotr24(T.te0(s2 & 0xff)) ^ k3;
    u32 m3 = T.te0(u3 >> 24) ^ rotr8(T.te0((u0 >> 16) & 0xff)) ^
             rotr16(T.te0((u1 >> 8) & 0xff)) ^ rotr24(T.te0(u2 & 0xff)) ^ k3;
    s0 = n0; s1 = n1; s2 = n2; s3 = n3;
    u0 = m0; u1 = m1; u2 = m2; u3 = m3;
  }
  // round-10 key
  {
    u32 w_ = (k3 << 8) | (k3 >> 24);
    w_ = (T.sbox((w_ >> 24) & 0xff) << 24) | (T.sbox((w_ >> 16) & 0xff) << 16) |
         (T.sbox((w_ >> 8) & 0xff) << 8) | T.sbox(w_ & 0xff);
    w_ ^= (rcon << 24);
    k0 ^= w_; k1 ^= k0; k2 ^= k1; k3 ^= k2;
  }
  u32 o0 = ((T.sbox(s0 >> 24) << 24) | (T.sbox((s1 >> 16) & 0xff) << 16) |
            (T.sbox((s2 >> 8) & 0xff) << 8) | T.sbox(s3 & 0xff)) ^ k0;
  u32 p0 = ((T.sbox(u0 >> 24) << 24) | (T.sbox((u1 >> 16) & 0xff) << 16) |
            (T.sbox((u2 >> 8) & 0xff) << 8) | T.sbox(u3 & 0xff)) ^ k0;
  if constexpr (LOW) {
    ra = make_uint4(__builtin_bswap32(o0), 0, 0, 0);
    rb = make_uint4(__builtin_bswap32(p0), 0, 0, 0);
    return;
  }
  u32 o1 = ((T.sbox(s1 >> 24) << 24) | (T.sbox((s2 >> 16) & 0xff) << 16) |
            (T.sbox((s3 >> 8) & 0xff) << 8) | T.sbox(s0 & 0xff)) ^ k1;
  u32 p1 = ((T.sbox(u1 >> 24) << 24) | (T.sbox((u2 >> 16) & 0xff) << 16) |
            (T.sbox((u3 >> 8) & 0xff) << 8) | T.sbox(u0 & 0xff)) ^ k1;
  u32 o2 = ((T.sbox(s2 >> 24) << 24) | (T.sbox((s3 >> 16) & 0xff) << 16) |
            (T.sbox((s0 >> 8) & 0xff) << 8) | T.sbox(s1 & 0xff)) ^ k2;
  u32 p2 = ((T.sbox(u2 >> 24) << 24) | (T.sbox((u3 >> 16) & 0xff) << 16) |
            (T.sbox((u0 >> 8) & 0xff) << 8) | T.sbox(u1 & 0xff)) ^ k2;
  u32 o3 = ((T.sbox(s3 >> 24) << 24) | (T.sbox((s0 >> 16) & 0xff) << 16) |
            (T.sbox((s1 >> 8) & 0xff) << 8) | T.sbox(s2 & 0xff)) ^ k3;
  u32 p3 = ((T.sbox(u3 >> 24) << 24) | (T.sbox((u0 >> 16) & 0xff) << 16) |
            (T.sbox((u1 >> 8) & 0xff) << 8) | T.sbox(u2 & 0xff)) ^ k3;
  ra = make_uint4(__builtin_bswap32(o0), __builtin_bswap32(o1),
                  __builtin_bswap32(o2), __builtin_bswap32(o3));
  rb = make_uint4(__builtin_bswap32(p0), __builtin_bswap32(p1),
                  __builtin_bswap32(p2), __builtin_bswap32(p3));
}

// ---------------------------------------------------------------------------
// PRF dispatch (fused kernel: AES uses the replicated LDS table)
// ---------------------------------------------------------------------------
template <int PRF>
__device__ __forceinline__ uint4 prf_full(uint4 seed, u32 pos,
                                          const AesLds& T) {
  if constexpr (PRF == PRF_DUMMY) return prf_dummy_full(seed, pos);
  if constexpr (PRF == PRF_SALSA20) return salsa12_core(seed, pos);
  if constexpr (PRF == PRF_CHACHA20) return chacha12_core(seed, pos);
  if constexpr (PRF == PRF_AES128) {
    uint4 a, b;
    aes_cipher_pair<false>(seed, T, a, b);
    return pos ? b : a;
  }
}

template <int PRF>
__device__ __forceinline__ void prf_pair(uint4 seed, const AesLds& T,
                                         uint4& r0, uint4& r1) {
  if constexpr (PRF == PRF_AES128) {
    aes_cipher_pair<false>(seed, T, r0, r1);
  } else if constexpr (PRF == PRF_SALSA20) {
    salsa12_pair_core(seed, r0, r1);
  } else if constexpr (PRF == PRF_CHACHA20) {
    chacha12_pair_core(seed, r0, r1);
  } else {
    r0 = prf_full<PRF>(seed, 0, T);
    r1 = prf_full<PRF>(seed, 1, T);
  }
}

template <int PRF>
__device__ __forceinline__ void prf_pair_low(uint4 seed, const AesLds& T,
                                             u32& r0, u32& r1) {
  if constexpr (PRF == PRF_DUMMY) {
    r0 = seed.x * 4242u + 4242u;
    r1 = seed.x * 4243u + 4243u;
  } else if constexpr (PRF == PRF_SALSA20) {
    salsa12_pair_low_core(seed, r0, r1);
  } else if constexpr (PRF == PRF_CHACHA20) {
    chacha12_pair_low_core(seed, r0, r1);
  } else {
    uint4 a, b;
    aes_cipher_pair<true>(seed, T, a, b);
    r0 = a.x;
    r1 = b.x;
  }
}

template <int PRF>
__device__ __forceinline__ void expand_pair(uint4 seed, int i,
                                            const uint4* cw_lds,
                                            const AesLds& T, uint4& c0,
                                            uint4& c1) {
  // Issue the correction-word reads BEFORE the PRF core: their address
  // depends only on the seed parity, and the ~600-instruction cipher
  // hides the LDS latency (left to itself the scheduler sinks them to
  // +5 instructions before their wait — measured in the .s).
  const int sel = (int)(seed.x & 1u);
  uint4 cw0 = cw_lds[sel * 64 + i * 2 + 0];
  uint4 cw1 = cw_lds[sel * 64 + i * 2 + 1];
  uint4 p0, p1;
  prf_pair<PRF>(seed, T, p0, p1);
  c0 = add128(p0, cw0);
  c1 = add128(p1, cw1);
}

// ---------------------------------------------------------------------------
// Main kernel.  LDS map (u32 granularity):
//   [cw: 128 uint4][shared: Z*max(2, DS-3) uint4][aes: 0/8192 u32][red]
// The `shared` region is the phase-1 ping-pong (2*Z uint4) first, then the
// cold DFS stack (levels 1..DS-3) — never live simultaneously.
// Hot stack levels DS-1 / DS-2 are the registers rtop1 / rtop2.
// ---------------------------------------------------------------------------
// slog: log2 of the DFS range split — each key's 2^(DS-1) leaf pairs are
// divided over 2^slog workgroups (j-split).  The split changes neither the
// table layout nor the per-thread subtree assignment: workgroup (key, seg)
// walks pairs [seg*P, (seg+1)*P) of every thread's subtree, paying one
// extra targeted descent (DS-1 pair expansions) to enter at seg*P.  Fused
// partials combine with wrapping u32 atomicAdd (exact mod 2^32).  This is
// what fills 256 CUs at small batch (batch 512 alone = only 2 WGs/CU) and
// doubles as the single-key low-latency mode (the reference needs a
// separate cooperative-groups kernel for that: dpf_coop.cu).
template <int PRF, bool FUSED>
__global__ __launch_bounds__(256) void dpf_eval_kernel(
    const int* __restrict__ keys, const u32* __restrict__ table,
    u32* __restrict__ out, const u32* __restrict__ aes_tabs,
    uint4* __restrict__ scratch, int depth, int zlog, int slog,
    long long n) {
  extern __shared__ u32 smem[];
  const int Z = 1 << zlog;
  const int DS = depth - zlog;  // subtree splits per thread (>= 1)
  // Sibling-stack placement by pop frequency (level d pops once per
  // 2^(DS-1-d) pairs): deepest two levels in registers, the next
  // MAX_LDS_LEVELS in LDS, the shallow remainder in global scratch
  // (touched <= 1/64 of steps) — keeping LDS small enough for 3+
  // workgroups per CU even with the 32 KB AES table resident.
  const int lds_levels =
      DS > 3 ? (DS - 3 < MAX_LDS_LEVELS ? DS - 3 : MAX_LDS_LEVELS) : 0;
  const int lds_base = DS - 2 - lds_levels;  // levels [lds_base, DS-3] in LDS
  const int glob_levels = (DS - 3) - lds_levels;  // levels [1, lds_base-1]
  const int t = (int)threadIdx.x;
  const int key_id = (int)(blockIdx.x >> slog);
  const int seg = (int)(blockIdx.x & ((1u << slog) - 1));
  const long long key_base = (long long)key_id * 524;

  uint4* cw_lds = reinterpret_cast<uint4*>(smem);   // 128 entries
  uint4* shared_region = cw_lds + 128;
  const int shared_u4 = Z * (lds_levels > 2 ? lds_levels : 2);
  uint4* gstack = scratch + (size_t)blockIdx.x * (glob_levels > 0 ? glob_levels : 0) * Z;
  u32* aes_lds = reinterpret_cast<u32*>(shared_region + shared_u4);
  u32* red = aes_lds + (PRF == PRF_AES128 ? AES_LDS_WORDS : 0);

  // Stage codewords; replicate the AES te0 table 32-way interleaved.
  for (int idx = t; idx < 128; idx += blockDim.x)
    cw_lds[idx] = reinterpret_cast<const uint4*>(keys + key_base + 4)[idx];
  if constexpr (PRF == PRF_AES128) {
    for (int e = t; e < 256; e += blockDim.x) {
      const u32 v = aes_tabs[e];  // global layout: te0 at offset 0
#pragma unroll
      for (int c = 0; c < AES_REP; ++c) aes_lds[e * AES_REP + c] = v;
    }
  }
  AesLds T{aes_lds, (u32)(t & (AES_REP - 1))};
  uint4* pp = shared_region;
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
      uint4 v = prf_full<PRF>(parent, (u32)(t & 1), T);
      const int sel = (int)(parent.x & 1u);
      b[t] = add128(v, cw_lds[sel * 64 + i * 2 + (t & 1)]);
    }
    __syncthreads();
    uint4* tmp = a;
    a = b;
    b = tmp;
  }
  uint4 cur = a[t];
  __syncthreads();  // everyone holds their frontier seed; pp is now free
  uint4* stack = shared_region;  // LDS stack levels [lds_base, DS-3]

  // Targeted descent to leaf-pair j_lo: at split d take bit (DS-1-d) of
  // j_lo, always recording the bit-1 child in the level's sibling slot
  // (when the bit is 1 the slot is stale, but it is provably rewritten by
  // a later descent before its next pop).
  const long long pairs = 1LL << (DS - 1);
  const long long j_lo = (long long)seg * (pairs >> slog);
  const long long j_hi = j_lo + (pairs >> slog);
  uint4 rtop1 = cur, rtop2 = cur;  // hot levels DS-1 / DS-2
  for (int d = 1; d <= DS - 1; ++d) {
    uint4 c0, c1;
    expand_pair<PRF>(cur, DS - d, cw_lds, T, c0, c1);
    if (d == DS - 1) rtop1 = c1;
    else if (d == DS - 2) rtop2 = c1;
    else if (d >= lds_base) stack[(d - lds_base) * Z + t] = c1;
    else gstack[(size_t)(d - 1) * Z + t] = c1;
    cur = ((j_lo >> (DS - 1 - d)) & 1) ? c1 : c0;
  }

  u32 acc[16];
#pragma unroll
  for (int w = 0; w < 16; ++w) acc[w] = 0;

  // leaf-level correction words (eval level 0) are read every iteration:
  // hoist their low words into registers
  const u32 cwl[2][2] = {{cw_lds[0].x, cw_lds[1].x},
                         {cw_lds[64].x, cw_lds[65].x}};

  for (long long j = j_lo; j < j_hi; ++j) {
    // Issue the two table-row loads first: the leaf ciphers below hide
    // their latency.
    uint4 r0q[4], r1q[4];
    if constexpr (FUSED) {
      const uint4* rows =
          reinterpret_cast<const uint4*>(table + ((j << (zlog + 1)) +
                                                  ((long long)t << 1)) * 16);
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        r0q[q] = rows[q];
        r1q[q] = rows[q + 4];
      }
    }

    u32 v0, v1;
    {
      u32 p0, p1;
      prf_pair_low<PRF>(cur, T, p0, p1);
      const int sel = (int)(cur.x & 1u);
      v0 = p0 + cwl[sel][0];
      v1 = p1 + cwl[sel][1];
    }

    if constexpr (FUSED) {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        acc[4 * q + 0] += v0 * r0q[q].x + v1 * r1q[q].x;
        acc[4 * q + 1] += v0 * r0q[q].y + v1 * r1q[q].y;
        acc[4 * q + 2] += v0 * r0q[q].z + v1 * r1q[q].z;
        acc[4 * q + 3] += v0 * r0q[q].w + v1 * r1q[q].w;
      }
    } else {
      u32* orow = out + (u64)key_id * (u64)n +
                  (u64)((j << (zlog + 1)) + ((long long)t << 1));
      orow[0] = v0;
      orow[1] = v1;
    }

    if (j + 1 == j_hi) break;
    // Pop the deepest pending sibling (register-resident on 3/4 of steps)
    // and descend its bit-0 spine.
    const int c = (int)(__ffsll((unsigned long long)(j + 1)) - 1);
    if (c == 0) {
      cur = rtop1;
      continue;  // next leaf-parent is the sibling itself
    }
    const int dpop = DS - 1 - c;
    cur = (c == 1) ? rtop2
          : (dpop >= lds_base) ? stack[(dpop - lds_base) * Z + t]
                               : gstack[(size_t)(dpop - 1) * Z + t];
    for (int d = dpop + 1; d <= DS - 1; ++d) {
      uint4 c0, c1;
      expand_pair<PRF>(cur, DS - d, cw_lds, T, c0, c1);
      if (d == DS - 1) rtop1 = c1;
      else if (d == DS - 2) rtop2 = c1;
      else if (d >= lds_base) stack[(d - lds_base) * Z + t] = c1;
      else gstack[(size_t)(d - 1) * Z + t] = c1;
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
      // Wrapping u32 atomic add combines the 2^slog segment partials
      // exactly (mod 2^32); `out` is zeroed by the caller.
      atomicAdd(&out[(u64)key_id * 16 + t], s);
    }
  }
}

// ---------------------------------------------------------------------------
// Naive oracle kernel: EvaluateFlat per (key, leaf), natural order.
// (Test-only; uses the flat 4-table AES layout at aes_tabs[0..1279]:
// te0|te1|te2|te3|sbox.)
// ---------------------------------------------------------------------------
__device__ __forceinline__ uint4 aes_flat(const u32* tabs, uint4 seed,
                                          u32 pos) {
  const u32* te0 = tabs;
  const u32* te1 = tabs + 256;
  const u32* te2 = tabs + 512;
  const u32* te3 = tabs + 768;
  const u32* sb = tabs + 1024;
  u32 rk[44];
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
  return make_uint4(__builtin_bswap32(o0), __builtin_bswap32(o1),
                    __builtin_bswap32(o2), __builtin_bswap32(o3));
}

template <int PRF>
__device__ __forceinline__ uint4 prf_naive(const u32* aes_lds, uint4 seed,
                                           u32 pos) {
  if constexpr (PRF == PRF_DUMMY) return prf_dummy_full(seed, pos);
  if constexpr (PRF == PRF_SALSA20) return salsa12_core(seed, pos);
  if constexpr (PRF == PRF_CHACHA20) return chacha12_core(seed, pos);
  if constexpr (PRF == PRF_AES128) return aes_flat(aes_lds, seed, pos);
}

template <int PRF>
__global__ __launch_bounds__(256) void dpf_naive_kernel(
    const int* __restrict__ keys, u32* __restrict__ out,
    const u32* __restrict__ aes_tabs, int depth, long long n) {
  extern __shared__ u32 smem[];
  uint4* cw_lds = reinterpret_cast<uint4*>(smem);
  u32* aes_lds = reinterpret_cast<u32*>(cw_lds + 128);
  const int t = (int)threadIdx.x;
  const long long key_base = (long long)blockIdx.x * 524;
  for (int idx = t; idx < 128; idx += blockDim.x)
    cw_lds[idx] = reinterpret_cast<const uint4*>(keys + key_base + 4)[idx];
  if constexpr (PRF == PRF_AES128) {
    for (int idx = t; idx < 1280; idx += blockDim.x)
      aes_lds[idx] = aes_tabs[idx];
  }
  __syncthreads();

  const int* rp = keys + key_base + 516;
  const long long leaf = (long long)blockIdx.y * blockDim.x + t;
  if (leaf >= n) return;
  uint4 key = make_uint4((u32)rp[0], (u32)rp[1], (u32)rp[2], (u32)rp[3]);
  long long rem = leaf;
  for (int i = depth - 1; i >= 0; --i) {
    const u32 bit = (u32)(rem & 1);
    uint4 v = prf_naive<PRF>(aes_lds, key, bit);
    const int sel = (int)(key.x & 1u);
    key = add128(v, cw_lds[sel * 64 + i * 2 + (int)bit]);
    rem >>= 1;
  }
  out[(u64)blockIdx.x * (u64)n + (u64)leaf] = key.x;
}

// ---------------------------------------------------------------------------
// PRF speed-of-light microbenchmark: dependent chain of pair expansions on
// register data only (no DFS, no table, no LDS traffic beyond the AES
// table).  Upper-bounds what any expansion kernel could reach; the fused
// kernel's pair rate divided by this is its efficiency.
// ---------------------------------------------------------------------------
template <int PRF>
__global__ __launch_bounds__(256) void prf_sol_kernel(
    const u32* __restrict__ aes_tabs, u32* __restrict__ out, int iters) {
  extern __shared__ u32 smem[];
  u32* aes_lds = smem;
  if constexpr (PRF == PRF_AES128) {
    for (int e = (int)threadIdx.x; e < 256; e += blockDim.x) {
      const u32 v = aes_tabs[e];
#pragma unroll
      for (int c = 0; c < AES_REP; ++c) aes_lds[e * AES_REP + c] = v;
    }
    __syncthreads();
  }
  AesLds T{aes_lds, (u32)(threadIdx.x & 31)};
  uint4 seed = make_uint4(threadIdx.x + 1, blockIdx.x + 2, 3, 4);
  for (int i = 0; i < iters; ++i) {
    uint4 r0, r1;
    prf_pair<PRF>(seed, T, r0, r1);
    seed = make_uint4(r0.x ^ r1.x, r0.y ^ r1.y, r0.z ^ r1.z, r0.w ^ r1.w);
  }
  out[(u64)blockIdx.x * blockDim.x + threadIdx.x] = seed.x;
}

void launch_prf_sol(std::uintptr_t aes_tabs, std::uintptr_t out, int blocks,
                    int iters, int prf, std::uintptr_t stream) {
  auto* a = reinterpret_cast<const u32*>(aes_tabs);
  auto* o = reinterpret_cast<u32*>(out);
  auto st = reinterpret_cast<hipStream_t>(stream);
  const size_t shmem = (prf == PRF_AES128) ? AES_LDS_WORDS * 4 : 0;
  switch (prf) {
    case PRF_DUMMY:
      hipLaunchKernelGGL(prf_sol_kernel<PRF_DUMMY>, dim3(blocks), dim3(256), shmem, st, a, o, iters);
      break;
    case PRF_SALSA20:
      hipLaunchKernelGGL(prf_sol_kernel<PRF_SALSA20>, dim3(blocks), dim3(256), shmem, st, a, o, iters);
      break;
    case PRF_CHACHA20:
      hipLaunchKernelGGL(prf_sol_kernel<PRF_CHACHA20>, dim3(blocks), dim3(256), shmem, st, a, o, iters);
      break;
    case PRF_AES128:
      hipLaunchKernelGGL(prf_sol_kernel<PRF_AES128>, dim3(blocks), dim3(256), shmem, st, a, o, iters);
      break;
    default:
      throw std::invalid_argument("unknown PRF");
  }
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Level-synchronized breadth-first expansion (the reference's
// dpf_breadth_first.cu:35-103 strategy, completing the strategy matrix
// next to naive / fused / two-stage).  One kernel launch per tree level;
// ping-pong u128 seed frontiers in global scratch; the last level writes
// low-32 one-hot shares in NATURAL index order with two coalesced
// streams (leaf idx = parent_pos | bit << (depth-1), because the eval
// recurrence consumes index bits LSB-first).
//
// This strategy pays n*16 B of frontier traffic per level pair and
// depth kernel launches — the fused DFS kernel exists precisely to avoid
// that; BFS is kept as a measured research point, not the production
// path.
// ---------------------------------------------------------------------------
template <int PRF>
__global__ __launch_bounds__(256) void dpf_bfs_level_kernel(
    const int* __restrict__ keys, const uint4* __restrict__ parents,
    uint4* __restrict__ children, u32* __restrict__ out,
    const u32* __restrict__ aes_tabs, int level, int depth, long long n) {
  extern __shared__ u32 smem[];
  u32* aes_lds = smem;
  uint4* cw_lvl = reinterpret_cast<uint4*>(
      smem + (PRF == PRF_AES128 ? AES_LDS_WORDS : 0));
  const int t = (int)threadIdx.x;
  const int key_id = (int)blockIdx.y;
  const long long key_base = (long long)key_id * 524;
  if constexpr (PRF == PRF_AES128) {
    for (int e = t; e < 256; e += blockDim.x) {
      const u32 v = aes_tabs[e];
#pragma unroll
      for (int c = 0; c < AES_REP; ++c) aes_lds[e * AES_REP + c] = v;
    }
  }
  // stage this level's 4 correction words (cw[sel][2i+b], i = eval level)
  const int i_eval = depth - 1 - level;
  if (t < 4) {
    const int sel = t >> 1, b = t & 1;
    cw_lvl[t] = reinterpret_cast<const uint4*>(keys + key_base + 4)
        [sel * 64 + i_eval * 2 + b];
  }
  __syncthreads();
  AesLds T{aes_lds, (u32)(t & (AES_REP - 1))};

  const long long n_parents = (long long)1 << level;
  const long long p = (long long)blockIdx.x * blockDim.x + t;
  if (p >= n_parents) return;

  uint4 seed;
  if (level == 0) {
    const int* rp = keys + key_base + 516;
    seed = make_uint4((u32)rp[0], (u32)rp[1], (u32)rp[2], (u32)rp[3]);
  } else {
    seed = parents[(u64)key_id * (n_parents) + (u64)p];
    // parents buffer is sized per level: indexed [key][p] with stride
    // n_parents (the launcher passes the matching base pointer)
  }
  const int sel = (int)(seed.x & 1u);
  uint4 c0, c1;
  prf_pair<PRF>(seed, T, c0, c1);
  c0 = add128(c0, cw_lvl[sel * 2 + 0]);
  c1 = add128(c1, cw_lvl[sel * 2 + 1]);
  if (level == depth - 1) {
    u32* orow = out + (u64)key_id * (u64)n;
    orow[p] = c0.x;
    orow[p + ((u64)1 << (depth - 1))] = c1.x;
  } else {
    uint4* crow = children + (u64)key_id * (n_parents * 2);
    crow[p] = c0;
    crow[p + n_parents] = c1;
  }
}

// ---------------------------------------------------------------------------
// Grid-wide cooperative single-key strategy (the reference's dpf_coop.cu:
// cudaLaunchCooperativeKernel + this_grid().sync() between tree levels,
// batch=1 only).  The whole grid walks one GGM tree breadth-first with a
// grid sync per level; with FUSED the leaves are MAC'd against the
// (leaf_perm-ordered) table into per-thread accumulators and combined
// with wrapping atomics — no second kernel.
//
// This exists for strategy-matrix completeness and as a measured
// comparison point: the production fused kernel's j-split serves the
// same single-key-latency role with NO grid-wide synchronization (its
// segments are independent), which is why it wins (see
// benchmarks/strategy_compare.py --coop).
// ---------------------------------------------------------------------------
#include <hip/hip_cooperative_groups.h>

template <int PRF, bool FUSED>
__global__ __launch_bounds__(256) void dpf_coop_kernel(
    const int* __restrict__ keys, const u32* __restrict__ table,
    u32* __restrict__ out, const u32* __restrict__ aes_tabs,
    uint4* __restrict__ ping, uint4* __restrict__ pong, int depth, int zlog,
    long long n) {
  extern __shared__ u32 smem[];
  u32* aes_lds = smem;
  const int t = (int)threadIdx.x;
  if constexpr (PRF == PRF_AES128) {
    for (int e = t; e < 256; e += blockDim.x) {
      const u32 v = aes_tabs[e];
#pragma unroll
      for (int c = 0; c < AES_REP; ++c) aes_lds[e * AES_REP + c] = v;
    }
    __syncthreads();
  }
  AesLds T{aes_lds, (u32)(t & (AES_REP - 1))};
  auto grid = cooperative_groups::this_grid();
  const long long gsize = (long long)gridDim.x * blockDim.x;
  const long long gtid = (long long)blockIdx.x * blockDim.x + t;
  const uint4* cw = reinterpret_cast<const uint4*>(keys + 4);

  u32 acc[16];
#pragma unroll
  for (int w = 0; w < 16; ++w) acc[w] = 0;

  for (int level = 0; level < depth; ++level) {
    const long long n_parents = (long long)1 << level;
    const int i_eval = depth - 1 - level;
    for (long long p = gtid; p < n_parents; p += gsize) {
      uint4 seed;
      if (level == 0) {
        const int* rp = keys + 516;
        seed = make_uint4((u32)rp[0], (u32)rp[1], (u32)rp[2], (u32)rp[3]);
      } else {
        seed = ping[p];
      }
      const int sel = (int)(seed.x & 1u);
      uint4 c0, c1;
      prf_pair<PRF>(seed, T, c0, c1);
      c0 = add128(c0, cw[sel * 64 + i_eval * 2 + 0]);
      c1 = add128(c1, cw[sel * 64 + i_eval * 2 + 1]);
      if (level < depth - 1) {
        pong[p] = c0;
        pong[p + n_parents] = c1;
      } else {
        // natural leaf indices: p and p | n/2 (bits consumed LSB-first)
        const long long i0 = p, i1 = p | (n >> 1);
        if constexpr (FUSED) {
#pragma unroll
          for (int q = 0; q < 2; ++q) {
            const long long idx = q ? i1 : i0;
            const u32 v = q ? c1.x : c0.x;
            // leaf_perm(idx) in-kernel (see dpf_core.cc layout contract)
            const int ds = depth - zlog;
            const u32 tl = (u32)(idx & ((1u << zlog) - 1));
            const u32 tpos = __brev(tl) >> (32 - zlog);
            const u32 jm = (u32)((idx >> zlog) & (((long long)1 << (ds - 1)) - 1));
            const u32 j = ds > 1 ? (__brev(jm) >> (33 - ds)) : 0;
            const u32 b = (u32)(idx >> (depth - 1));
            const long long row = ((long long)j << (zlog + 1)) |
                                  ((long long)tpos << 1) | b;
            const u32* trow = table + (u64)row * 16;
#pragma unroll
            for (int w = 0; w < 16; ++w) acc[w] += v * trow[w];
          }
        } else {
          out[i0] = c0.x;
          out[i1] = c1.x;
        }
      }
    }
    if (level < depth - 1) {
      grid.sync();
      uint4* tmp = ping;
      ping = pong;
      pong = tmp;
    }
  }
  if constexpr (FUSED) {
#pragma unroll
    for (int w = 0; w < 16; ++w)
      if (acc[w]) atomicAdd(out + w, acc[w]);
  }
}

// ---------------------------------------------------------------------------
// ALU / PRF probe kernels — the unit-test analog of the reference's
// dpf_gpu/tests/test_128_bit.cu:192-200 (device add/mul/pack asserted
// against host __int128): one element per thread, results compared
// elementwise against the CPU core in tests/test_gpu_alu.py.  These probe
// the EXACT device functions the production kernels inline (add128,
// prf_pair, prf_pair_low, prf_full with the replicated-LDS AES table), so
// a PRF-core bug fails a unit test, not only end-to-end reconstruction.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void probe_alu_kernel(
    const uint4* __restrict__ a, const uint4* __restrict__ b,
    uint4* __restrict__ add_out, uint4* __restrict__ mul_out, int count) {
  const int i = (int)(blockIdx.x * blockDim.x + threadIdx.x);
  if (i >= count) return;
  add_out[i] = add128(a[i], b[i]);
  const unsigned __int128 x =
      ((unsigned __int128)(((u64)a[i].w << 32) | a[i].z) << 64) |
      (((u64)a[i].y << 32) | a[i].x);
  const unsigned __int128 y =
      ((unsigned __int128)(((u64)b[i].w << 32) | b[i].z) << 64) |
      (((u64)b[i].y << 32) | b[i].x);
  const unsigned __int128 r = x * y;
  const u64 lo = (u64)r, hi = (u64)(r >> 64);
  mul_out[i] = make_uint4((u32)lo, (u32)(lo >> 32), (u32)hi, (u32)(hi >> 32));
}

template <int PRF>
__global__ __launch_bounds__(256) void probe_prf_kernel(
    const uint4* __restrict__ seeds, const u32* __restrict__ aes_tabs,
    uint4* __restrict__ pair0, uint4* __restrict__ pair1,
    uint4* __restrict__ single0, uint4* __restrict__ single1,
    u32* __restrict__ low0, u32* __restrict__ low1, int count) {
  extern __shared__ u32 smem[];
  u32* aes_lds = smem;
  if constexpr (PRF == PRF_AES128) {
    for (int e = (int)threadIdx.x; e < 256; e += blockDim.x) {
      const u32 v = aes_tabs[e];
#pragma unroll
      for (int c = 0; c < AES_REP; ++c) aes_lds[e * AES_REP + c] = v;
    }
    __syncthreads();
  }
  AesLds T{aes_lds, (u32)(threadIdx.x & (AES_REP - 1))};
  const int i = (int)(blockIdx.x * blockDim.x + threadIdx.x);
  if (i >= count) return;
  const uint4 s = seeds[i];
  uint4 r0, r1;
  prf_pair<PRF>(s, T, r0, r1);
  pair0[i] = r0;
  pair1[i] = r1;
  single0[i] = prf_full<PRF>(s, 0, T);
  single1[i] = prf_full<PRF>(s, 1, T);
  u32 l0, l1;
  prf_pair_low<PRF>(s, T, l0, l1);
  low0[i] = l0;
  low1[i] = l1;
}

void launch_probe_alu(std::uintptr_t a, std::uintptr_t b,
                      std::uintptr_t add_out, std::uintptr_t mul_out,
                      int count, std::uintptr_t stream) {
  auto st = reinterpret_cast<hipStream_t>(stream);
  const int blocks = (count + 255) / 256;
  hipLaunchKernelGGL(probe_alu_kernel, dim3(blocks), dim3(256), 0, st,
                     reinterpret_cast<const uint4*>(a),
                     reinterpret_cast<const uint4*>(b),
                     reinterpret_cast<uint4*>(add_out),
                     reinterpret_cast<uint4*>(mul_out), count);
  HIP_CHECK(hipGetLastError());
}

void launch_probe_prf(std::uintptr_t seeds, std::uintptr_t aes_tabs,
                      std::uintptr_t pair0, std::uintptr_t pair1,
                      std::uintptr_t single0, std::uintptr_t single1,
                      std::uintptr_t low0, std::uintptr_t low1, int count,
                      int prf, std::uintptr_t stream) {
  auto st = reinterpret_cast<hipStream_t>(stream);
  const int blocks = (count + 255) / 256;
  const size_t shmem = (prf == PRF_AES128) ? AES_LDS_WORDS * 4 : 0;
  auto* sp = reinterpret_cast<const uint4*>(seeds);
  auto* ap = reinterpret_cast<const u32*>(aes_tabs);
  auto* p0 = reinterpret_cast<uint4*>(pair0);
  auto* p1 = reinterpret_cast<uint4*>(pair1);
  auto* s0 = reinterpret_cast<uint4*>(single0);
  auto* s1 = reinterpret_cast<uint4*>(single1);
  auto* l0 = reinterpret_cast<u32*>(low0);
  auto* l1 = reinterpret_cast<u32*>(low1);
  switch (prf) {
    case PRF_DUMMY:
      hipLaunchKernelGGL(probe_prf_kernel<PRF_DUMMY>, dim3(blocks), dim3(256),
                         shmem, st, sp, ap, p0, p1, s0, s1, l0, l1, count);
      break;
    case PRF_SALSA20:
      hipLaunchKernelGGL(probe_prf_kernel<PRF_SALSA20>, dim3(blocks),
                         dim3(256), shmem, st, sp, ap, p0, p1, s0, s1, l0, l1,
                         count);
      break;
    case PRF_CHACHA20:
      hipLaunchKernelGGL(probe_prf_kernel<PRF_CHACHA20>, dim3(blocks),
                         dim3(256), shmem, st, sp, ap, p0, p1, s0, s1, l0, l1,
                         count);
      break;
    case PRF_AES128:
      hipLaunchKernelGGL(probe_prf_kernel<PRF_AES128>, dim3(blocks), dim3(256),
                         shmem, st, sp, ap, p0, p1, s0, s1, l0, l1, count);
      break;
    default:
      throw std::invalid_argument("unknown PRF");
  }
  HIP_CHECK(hipGetLastError());
}

namespace {
uint4* get_scratch(size_t bytes, hipStream_t stream);  // defined below

template <int PRF, bool FUSED>
void launch_coop_t(const int* keys, const u32* table, u32* out,
                   const u32* aes_tabs, uint4* ping, uint4* pong, int depth,
                   int zlog, long long n, hipStream_t st) {
  auto kern = dpf_coop_kernel<PRF, FUSED>;
  const size_t shmem = (PRF == PRF_AES128) ? AES_LDS_WORDS * 4 : 0;
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  int coop = 0;
  HIP_CHECK(hipDeviceGetAttribute(&coop, hipDeviceAttributeCooperativeLaunch,
                                  dev));
  if (!coop)
    throw std::runtime_error("device does not support cooperative launch");
  int num_cu = 0;
  HIP_CHECK(hipDeviceGetAttribute(&num_cu,
                                  hipDeviceAttributeMultiprocessorCount, dev));
  int max_blocks = 0;
  HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &max_blocks, (const void*)kern, 256, shmem));
  long long grid = (long long)num_cu * max_blocks;
  const long long need = ((n / 2) + 255) / 256;  // widest level
  if (grid > need) grid = need;
  if (grid < 1) grid = 1;
  void* args[] = {(void*)&keys, (void*)&table, (void*)&out,
                  (void*)&aes_tabs, (void*)&ping, (void*)&pong,
                  (void*)&depth, (void*)&zlog, (void*)&n};
  HIP_CHECK(hipLaunchCooperativeKernel((const void*)kern,
                                       dim3((unsigned)grid), dim3(256), args,
                                       shmem, st));
}
}  // namespace

void launch_coop(std::uintptr_t keys, std::uintptr_t table, std::uintptr_t out,
                 std::uintptr_t aes_tabs, long long n, int depth, int zlog,
                 int prf, bool fused, std::uintptr_t stream) {
  if (depth < 1 || ((long long)1 << depth) != n)
    throw std::invalid_argument("bad depth/n");
  auto st = reinterpret_cast<hipStream_t>(stream);
  const size_t half = (size_t)(n / 2) * sizeof(uint4);
  uint4* ping = get_scratch(2 * half > 0 ? 2 * half : 32, st);
  uint4* pong = ping + (size_t)(n / 2);
  auto* kp = reinterpret_cast<const int*>(keys);
  auto* tp = reinterpret_cast<const u32*>(table);
  auto* op = reinterpret_cast<u32*>(out);
  auto* ap = reinterpret_cast<const u32*>(aes_tabs);
#define COOP_CASE(P)                                                       \
  case P:                                                                  \
    if (fused)                                                             \
      launch_coop_t<P, true>(kp, tp, op, ap, ping, pong, depth, zlog, n,  \
                             st);                                          \
    else                                                                   \
      launch_coop_t<P, false>(kp, tp, op, ap, ping, pong, depth, zlog, n, \
                              st);                                         \
    break;
  switch (prf) {
    COOP_CASE(PRF_DUMMY)
    COOP_CASE(PRF_SALSA20)
    COOP_CASE(PRF_CHACHA20)
    COOP_CASE(PRF_AES128)
    default:
      throw std::invalid_argument("unknown PRF");
  }
#undef COOP_CASE
}

namespace {
template <int PRF>
void launch_bfs_t(const int* keys, u32* out, const u32* aes_tabs,
                  uint4* ping, uint4* pong, int batch, long long n, int depth,
                  hipStream_t st) {
  const size_t shmem =
      (PRF == PRF_AES128 ? AES_LDS_WORDS * 4 : 0) + 4 * sizeof(uint4);
  for (int level = 0; level < depth; ++level) {
    const long long n_parents = (long long)1 << level;
    const long long blocks = (n_parents + 255) / 256;
    const uint4* parents = (level & 1) ? pong : ping;
    uint4* children = (level & 1) ? ping : pong;
    hipLaunchKernelGGL(dpf_bfs_level_kernel<PRF>,
                       dim3((unsigned)blocks, (unsigned)batch), dim3(256),
                       shmem, st, keys, parents, children, out, aes_tabs,
                       level, depth, n);
    HIP_CHECK(hipGetLastError());
  }
}
}  // namespace

void launch_bfs(std::uintptr_t keys, std::uintptr_t out,
                std::uintptr_t aes_tabs, int batch, long long n, int depth,
                int prf, std::uintptr_t stream) {
  if (batch <= 0) return;
  if (depth < 1 || ((long long)1 << depth) != n)
    throw std::invalid_argument("bad depth/n");
  auto st = reinterpret_cast<hipStream_t>(stream);
  // two ping-pong frontier buffers, each batch * n/2 seeds
  const size_t half = (size_t)batch * (size_t)(n / 2) * sizeof(uint4);
  if (2 * half > (size_t)48 << 30)
    throw std::invalid_argument("BFS frontier would exceed 48 GiB scratch");
  uint4* ping = get_scratch(2 * half > 0 ? 2 * half : 16, st);
  uint4* pong = ping + (size_t)batch * (size_t)(n / 2);
  auto* kp = reinterpret_cast<const int*>(keys);
  auto* op = reinterpret_cast<u32*>(out);
  auto* ap = reinterpret_cast<const u32*>(aes_tabs);
  switch (prf) {
    case PRF_DUMMY:
      launch_bfs_t<PRF_DUMMY>(kp, op, ap, ping, pong, batch, n, depth, st);
      break;
    case PRF_SALSA20:
      launch_bfs_t<PRF_SALSA20>(kp, op, ap, ping, pong, batch, n, depth, st);
      break;
    case PRF_CHACHA20:
      launch_bfs_t<PRF_CHACHA20>(kp, op, ap, ping, pong, batch, n, depth, st);
      break;
    case PRF_AES128:
      launch_bfs_t<PRF_AES128>(kp, op, ap, ping, pong, batch, n, depth, st);
      break;
    default:
      throw std::invalid_argument("unknown PRF");
  }
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------
namespace {

size_t fused_shmem_bytes(int Z, int DS, int prf) {
  int lds_levels = DS > 3 ? DS - 3 : 0;
  if (lds_levels > MAX_LDS_LEVELS) lds_levels = MAX_LDS_LEVELS;
  size_t bytes = 128 * 16;                                  // codewords
  bytes += (size_t)Z * (lds_levels > 2 ? lds_levels : 2) * 16;  // pp/stack
  if (prf == PRF_AES128) bytes += AES_LDS_WORDS * 4;        // replicated te0
  bytes += (size_t)(Z / 64) * 16 * 4;                       // reduction
  return bytes;
}

// Grow-only per-device scratch for the global stack levels.  Reused
// across launches; sized for the largest request seen.  NOTE: concurrent
// fused/expand launches on DIFFERENT streams of one device would share
// this buffer — the python API serializes launches per DPF call, which
// is the supported pattern.
//
// Outgrown buffers are RETIRED, never freed: a hipGraph captured while a
// launch used the old buffer holds its raw pointer and replays against it
// (round-1 advisor finding — freeing here turned later replays into
// use-after-free).  Growth is geometric (2x), so retired memory is bounded
// by the live buffer's size and the retired list stays O(log largest).
std::mutex g_scratch_mu;
void* g_scratch[64] = {};
size_t g_scratch_bytes[64] = {};
std::vector<void*> g_scratch_retired[64];

uint4* get_scratch(size_t bytes, hipStream_t /*stream*/) {
  if (bytes == 0) return nullptr;
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  std::lock_guard<std::mutex> lock(g_scratch_mu);
  if (g_scratch_bytes[dev] < bytes) {
    if (g_scratch[dev]) g_scratch_retired[dev].push_back(g_scratch[dev]);
    size_t want = g_scratch_bytes[dev] ? g_scratch_bytes[dev] : bytes;
    while (want < bytes) want *= 2;
    HIP_CHECK(hipMalloc(&g_scratch[dev], want));
    g_scratch_bytes[dev] = want;
  }
  return reinterpret_cast<uint4*>(g_scratch[dev]);
}

template <int PRF, bool FUSED>
void launch_eval_t(const int* keys, const u32* table, u32* out,
                   const u32* aes_tabs, int batch, long long n, int depth,
                   int zlog, hipStream_t stream) {
  const int Z = 1 << zlog;
  const int DS = depth - zlog;
  // j-split: grow the grid to >= 1024 workgroups (4 per CU) so small
  // batches still fill the chip; each extra split costs one targeted
  // descent (DS-1 pair expansions) per workgroup — negligible against
  // pairs/2^slog leaf pairs.
  int slog = 0;
  while ((batch << (slog + 1)) <= 2048 && slog + 2 <= DS - 1 && slog < 8)
    ++slog;
  const size_t shmem = fused_shmem_bytes(Z, DS, PRF);
  auto kern = dpf_eval_kernel<PRF, FUSED>;
  if (shmem > 65536) {
    HIP_CHECK(hipFuncSetAttribute((const void*)kern,
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  (int)shmem));
  }
  int lds_levels = DS > 3 ? DS - 3 : 0;
  if (lds_levels > MAX_LDS_LEVELS) lds_levels = MAX_LDS_LEVELS;
  const int glob_levels = (DS > 3 ? DS - 3 : 0) - lds_levels;
  uint4* scratch = get_scratch(
      (size_t)(batch << slog) * glob_levels * Z * 16, stream);
  hipLaunchKernelGGL(kern, dim3((unsigned)(batch << slog)), dim3((unsigned)Z),
                     shmem, stream, keys, table, out, aes_tabs, scratch,
                     depth, zlog, slog, n);
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
