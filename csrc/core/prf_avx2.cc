// AVX2 8-lane Salsa20/12 and ChaCha20/12 for the CPU expansion hot loop:
// eight independent seeds per call, one state word per __m256i lane.
// Bit-exact with the scalar implementations in prf.cc (same spec, same
// i/o conventions); runtime-guarded.

#include "dpf_core.h"

#if defined(__AVX2__)
#include <immintrin.h>

namespace gpudpf {

bool avx2_available() {
  static const bool ok = __builtin_cpu_supports("avx2");
  return ok;
}

namespace {

inline __m256i rotl(__m256i v, int s) {
  return _mm256_or_si256(_mm256_slli_epi32(v, s), _mm256_srli_epi32(v, 32 - s));
}

// gather word w (0 = low) of each of 8 u128 seeds into one vector
inline __m256i seed_word(const u128* seeds, int w) {
  alignas(32) u32 tmp[8];
  for (int i = 0; i < 8; ++i) tmp[i] = (u32)(seeds[i] >> (32 * w));
  return _mm256_load_si256(reinterpret_cast<const __m256i*>(tmp));
}

inline void store_out(u128* out, __m256i hi3, __m256i hi2, __m256i hi1,
                      __m256i lo) {
  alignas(32) u32 a[8], b[8], c[8], d[8];
  _mm256_store_si256(reinterpret_cast<__m256i*>(a), hi3);
  _mm256_store_si256(reinterpret_cast<__m256i*>(b), hi2);
  _mm256_store_si256(reinterpret_cast<__m256i*>(c), hi1);
  _mm256_store_si256(reinterpret_cast<__m256i*>(d), lo);
  for (int i = 0; i < 8; ++i)
    out[i] = ((u128)a[i] << 96) | ((u128)b[i] << 64) | ((u128)c[i] << 32) |
             (u128)d[i];
}

#define SQR(a, b, c, d)            \
  b = _mm256_xor_si256(b, rotl(_mm256_add_epi32(a, d), 7));  \
  c = _mm256_xor_si256(c, rotl(_mm256_add_epi32(b, a), 9));  \
  d = _mm256_xor_si256(d, rotl(_mm256_add_epi32(c, b), 13)); \
  a = _mm256_xor_si256(a, rotl(_mm256_add_epi32(d, c), 18))

#define CQR(a, b, c, d)                                       \
  a = _mm256_add_epi32(a, b); d = _mm256_xor_si256(d, a); d = rotl(d, 16); \
  c = _mm256_add_epi32(c, d); b = _mm256_xor_si256(b, c); b = rotl(b, 12); \
  a = _mm256_add_epi32(a, b); d = _mm256_xor_si256(d, a); d = rotl(d, 8);  \
  c = _mm256_add_epi32(c, d); b = _mm256_xor_si256(b, c); b = rotl(b, 7)

}  // namespace

void salsa12_x8(const u128* seeds, u32 pos, u128* out) {
  const __m256i c0 = _mm256_set1_epi32(0x65787061);
  const __m256i c5 = _mm256_set1_epi32(0x6e642033);
  const __m256i c10 = _mm256_set1_epi32(0x322d6279);
  const __m256i c15 = _mm256_set1_epi32(0x7465206b);
  const __m256i zero = _mm256_setzero_si256();
  __m256i i1 = seed_word(seeds, 3), i2 = seed_word(seeds, 2),
          i3 = seed_word(seeds, 1), i4 = seed_word(seeds, 0);
  __m256i x0 = c0, x1 = i1, x2 = i2, x3 = i3, x4 = i4, x5 = c5, x6 = zero,
          x7 = zero, x8 = zero, x9 = _mm256_set1_epi32((int)pos), x10 = c10,
          x11 = zero, x12 = zero, x13 = zero, x14 = zero, x15 = c15;
  for (int r = 0; r < 6; ++r) {
    SQR(x0, x4, x8, x12);
    SQR(x5, x9, x13, x1);
    SQR(x10, x14, x2, x6);
    SQR(x15, x3, x7, x11);
    SQR(x0, x1, x2, x3);
    SQR(x5, x6, x7, x4);
    SQR(x10, x11, x8, x9);
    SQR(x15, x12, x13, x14);
  }
  store_out(out, _mm256_add_epi32(x1, i1), _mm256_add_epi32(x2, i2),
            _mm256_add_epi32(x3, i3), _mm256_add_epi32(x4, i4));
}

void chacha12_x8(const u128* seeds, u32 pos, u128* out) {
  const __m256i k0 = _mm256_set1_epi32(0x65787061);
  const __m256i k1 = _mm256_set1_epi32(0x6e642033);
  const __m256i k2 = _mm256_set1_epi32(0x322d6279);
  const __m256i k3 = _mm256_set1_epi32(0x7465206b);
  const __m256i zero = _mm256_setzero_si256();
  __m256i i4 = seed_word(seeds, 3), i5 = seed_word(seeds, 2),
          i6 = seed_word(seeds, 1), i7 = seed_word(seeds, 0);
  __m256i x0 = k0, x1 = k1, x2 = k2, x3 = k3, x4 = i4, x5 = i5, x6 = i6,
          x7 = i7, x8 = zero, x9 = zero, x10 = zero, x11 = zero, x12 = zero,
          x13 = _mm256_set1_epi32((int)pos), x14 = zero, x15 = zero;
  for (int r = 0; r < 6; ++r) {
    CQR(x0, x4, x8, x12);
    CQR(x1, x5, x9, x13);
    CQR(x2, x6, x10, x14);
    CQR(x3, x7, x11, x15);
    CQR(x0, x5, x10, x15);
    CQR(x1, x6, x11, x12);
    CQR(x2, x7, x8, x13);
    CQR(x3, x4, x9, x14);
  }
  store_out(out, _mm256_add_epi32(x4, i4), _mm256_add_epi32(x5, i5),
            _mm256_add_epi32(x6, i6), _mm256_add_epi32(x7, i7));
}

}  // namespace gpudpf
#else
namespace gpudpf {
bool avx2_available() { return false; }
void salsa12_x8(const u128*, u32, u128*) {}
void chacha12_x8(const u128*, u32, u128*) {}
}  // namespace gpudpf
#endif
