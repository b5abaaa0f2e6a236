// Standard AES-128 (FIPS-197) single-block encryption, written from the
// specification.  Tables (S-box and round T-tables) are generated
// programmatically from GF(2^8) arithmetic at first use rather than
// embedded as literals.  Verified against the FIPS-197 Appendix B vector in
// tests/test_prf_vectors.py and against the reference repo's behaviour
// (reference CPU AES: /root/reference/dpf_base/aes_core.h — an
// OpenSSL-derived implementation of the same standard cipher).

#include "dpf_core.h"

#include <mutex>

namespace gpudpf {
namespace {

struct AesTables {
  unsigned char sbox[256];
  u32 te0[256], te1[256], te2[256], te3[256];
};

unsigned char gf_mul(unsigned char a, unsigned char b) {
  unsigned char p = 0;
  for (int i = 0; i < 8; ++i) {
    if (b & 1) p ^= a;
    unsigned char hi = a & 0x80;
    a <<= 1;
    if (hi) a ^= 0x1b;
    b >>= 1;
  }
  return p;
}

const AesTables& tables() {
  static AesTables t;
  static std::once_flag once;
  std::call_once(once, [] {
    // S-box: multiplicative inverse in GF(2^8) followed by the affine map.
    // Build inverses via exhaustive product search (256^2, one-time cost).
    unsigned char inv[256] = {0};
    for (int a = 1; a < 256; ++a) {
      for (int b = 1; b < 256; ++b) {
        if (gf_mul((unsigned char)a, (unsigned char)b) == 1) {
          inv[a] = (unsigned char)b;
          break;
        }
      }
    }
    for (int x = 0; x < 256; ++x) {
      unsigned char q = inv[x];
      unsigned char s = 0x63;
      for (int i = 0; i < 8; ++i) {
        unsigned char bit =
            ((q >> i) ^ (q >> ((i + 4) & 7)) ^ (q >> ((i + 5) & 7)) ^
             (q >> ((i + 6) & 7)) ^ (q >> ((i + 7) & 7))) & 1;
        s ^= (unsigned char)(bit << i);
      }
      t.sbox[x] = s;
    }
    // T-tables: column transform of a single S-boxed byte.  With the state
    // column packed big-endian into a u32 (byte 0 = bits 31..24), the
    // MixColumns matrix row order gives:
    //   te0[x] = [2s, s, s, 3s]  (s = sbox[x])
    for (int x = 0; x < 256; ++x) {
      unsigned char s = t.sbox[x];
      unsigned char s2 = gf_mul(s, 2), s3 = gf_mul(s, 3);
      t.te0[x] = ((u32)s2 << 24) | ((u32)s << 16) | ((u32)s << 8) | (u32)s3;
      t.te1[x] = ((u32)s3 << 24) | ((u32)s2 << 16) | ((u32)s << 8) | (u32)s;
      t.te2[x] = ((u32)s << 24) | ((u32)s3 << 16) | ((u32)s2 << 8) | (u32)s;
      t.te3[x] = ((u32)s << 24) | ((u32)s << 16) | ((u32)s3 << 8) | (u32)s2;
    }
  });
  return t;
}

}  // namespace

void aes128_tables(u32 te0[256], u32 te1[256], u32 te2[256], u32 te3[256],
                   u32 sbox[256]) {
  const AesTables& t = tables();
  for (int i = 0; i < 256; ++i) {
    te0[i] = t.te0[i];
    te1[i] = t.te1[i];
    te2[i] = t.te2[i];
    te3[i] = t.te3[i];
    sbox[i] = t.sbox[i];
  }
}

void aes128_expand_key(const unsigned char key[16], u32 rk[44]) {
  const AesTables& t = tables();
  for (int i = 0; i < 4; ++i) {
    rk[i] = ((u32)key[4 * i] << 24) | ((u32)key[4 * i + 1] << 16) |
            ((u32)key[4 * i + 2] << 8) | (u32)key[4 * i + 3];
  }
  u32 rcon = 0x01000000u;
  for (int i = 4; i < 44; ++i) {
    u32 w = rk[i - 1];
    if (i % 4 == 0) {
      // RotWord + SubWord + Rcon
      w = (w << 8) | (w >> 24);
      w = ((u32)t.sbox[(w >> 24) & 0xff] << 24) |
          ((u32)t.sbox[(w >> 16) & 0xff] << 16) |
          ((u32)t.sbox[(w >> 8) & 0xff] << 8) | (u32)t.sbox[w & 0xff];
      w ^= rcon;
      rcon = (u32)gf_mul((unsigned char)(rcon >> 24), 2) << 24;
    }
    rk[i] = rk[i - 4] ^ w;
  }
}

void aes128_encrypt_block_rk(const u32 rk[44], const unsigned char in[16],
                             unsigned char out[16]) {
  const AesTables& t = tables();
  u32 s0 = (((u32)in[0] << 24) | ((u32)in[1] << 16) | ((u32)in[2] << 8) |
            (u32)in[3]) ^ rk[0];
  u32 s1 = (((u32)in[4] << 24) | ((u32)in[5] << 16) | ((u32)in[6] << 8) |
            (u32)in[7]) ^ rk[1];
  u32 s2 = (((u32)in[8] << 24) | ((u32)in[9] << 16) | ((u32)in[10] << 8) |
            (u32)in[11]) ^ rk[2];
  u32 s3 = (((u32)in[12] << 24) | ((u32)in[13] << 16) | ((u32)in[14] << 8) |
            (u32)in[15]) ^ rk[3];
  for (int r = 1; r < 10; ++r) {
    u32 n0 = t.te0[s0 >> 24] ^ t.te1[(s1 >> 16) & 0xff] ^
             t.te2[(s2 >> 8) & 0xff] ^ t.te3[s3 & 0xff] ^ rk[4 * r];
    u32 n1 = t.te0[s1 >> 24] ^ t.te1[(s2 >> 16) & 0xff] ^
             t.te2[(s3 >> 8) & 0xff] ^ t.te3[s0 & 0xff] ^ rk[4 * r + 1];
    u32 n2 = t.te0[s2 >> 24] ^ t.te1[(s3 >> 16) & 0xff] ^
             t.te2[(s0 >> 8) & 0xff] ^ t.te3[s1 & 0xff] ^ rk[4 * r + 2];
    u32 n3 = t.te0[s3 >> 24] ^ t.te1[(s0 >> 16) & 0xff] ^
             t.te2[(s1 >> 8) & 0xff] ^ t.te3[s2 & 0xff] ^ rk[4 * r + 3];
    s0 = n0; s1 = n1; s2 = n2; s3 = n3;
  }
  // Final round: SubBytes + ShiftRows + AddRoundKey (no MixColumns).
  u32 o0 = (((u32)t.sbox[s0 >> 24] << 24) |
            ((u32)t.sbox[(s1 >> 16) & 0xff] << 16) |
            ((u32)t.sbox[(s2 >> 8) & 0xff] << 8) | (u32)t.sbox[s3 & 0xff]) ^
           rk[40];
  u32 o1 = (((u32)t.sbox[s1 >> 24] << 24) |
            ((u32)t.sbox[(s2 >> 16) & 0xff] << 16) |
            ((u32)t.sbox[(s3 >> 8) & 0xff] << 8) | (u32)t.sbox[s0 & 0xff]) ^
           rk[41];
  u32 o2 = (((u32)t.sbox[s2 >> 24] << 24) |
            ((u32)t.sbox[(s3 >> 16) & 0xff] << 16) |
            ((u32)t.sbox[(s0 >> 8) & 0xff] << 8) | (u32)t.sbox[s1 & 0xff]) ^
           rk[42];
  u32 o3 = (((u32)t.sbox[s3 >> 24] << 24) |
            ((u32)t.sbox[(s0 >> 16) & 0xff] << 16) |
            ((u32)t.sbox[(s1 >> 8) & 0xff] << 8) | (u32)t.sbox[s2 & 0xff]) ^
           rk[43];
  u32 o[4] = {o0, o1, o2, o3};
  for (int i = 0; i < 4; ++i) {
    out[4 * i] = (unsigned char)(o[i] >> 24);
    out[4 * i + 1] = (unsigned char)(o[i] >> 16);
    out[4 * i + 2] = (unsigned char)(o[i] >> 8);
    out[4 * i + 3] = (unsigned char)o[i];
  }
}

void aes128_encrypt_block(const unsigned char key[16], const unsigned char in[16],
                          unsigned char out[16]) {
  u32 rk[44];
  aes128_expand_key(key, rk);
  aes128_encrypt_block_rk(rk, in, out);
}

}  // namespace gpudpf

// ---------------------------------------------------------------------------
// AES-NI fast path (x86).  Bit-exact with the table implementation above
// (both are FIPS-197 AES-128 on the same byte order); used by the hot CPU
// expansion loop when the host supports it.
// ---------------------------------------------------------------------------
#if defined(__AES__) && defined(__SSE4_1__)
#include <immintrin.h>

namespace gpudpf {

bool aesni_available() {
  static const bool ok = __builtin_cpu_supports("aes");
  return ok;
}

namespace {
inline __m128i ks_step(__m128i key, __m128i kg) {
  kg = _mm_shuffle_epi32(kg, _MM_SHUFFLE(3, 3, 3, 3));
  key = _mm_xor_si128(key, _mm_slli_si128(key, 4));
  key = _mm_xor_si128(key, _mm_slli_si128(key, 4));
  key = _mm_xor_si128(key, _mm_slli_si128(key, 4));
  return _mm_xor_si128(key, kg);
}
}  // namespace

void aes128_expand_key_ni(const unsigned char key[16], AesNiRoundKeys& rk) {
  __m128i* k = reinterpret_cast<__m128i*>(rk.rk);
  k[0] = _mm_loadu_si128(reinterpret_cast<const __m128i*>(key));
  k[1] = ks_step(k[0], _mm_aeskeygenassist_si128(k[0], 0x01));
  k[2] = ks_step(k[1], _mm_aeskeygenassist_si128(k[1], 0x02));
  k[3] = ks_step(k[2], _mm_aeskeygenassist_si128(k[2], 0x04));
  k[4] = ks_step(k[3], _mm_aeskeygenassist_si128(k[3], 0x08));
  k[5] = ks_step(k[4], _mm_aeskeygenassist_si128(k[4], 0x10));
  k[6] = ks_step(k[5], _mm_aeskeygenassist_si128(k[5], 0x20));
  k[7] = ks_step(k[6], _mm_aeskeygenassist_si128(k[6], 0x40));
  k[8] = ks_step(k[7], _mm_aeskeygenassist_si128(k[7], 0x80));
  k[9] = ks_step(k[8], _mm_aeskeygenassist_si128(k[8], 0x1b));
  k[10] = ks_step(k[9], _mm_aeskeygenassist_si128(k[9], 0x36));
}

// Encrypt the two counter blocks (0 and 1) under one schedule, chains
// interleaved.
void aes128_encrypt2_ni(const AesNiRoundKeys& rk, unsigned char out0[16],
                        unsigned char out1[16]) {
  const __m128i* k = reinterpret_cast<const __m128i*>(rk.rk);
  __m128i s0 = k[0];                                   // pt = 0
  __m128i s1 = _mm_xor_si128(_mm_cvtsi32_si128(1), k[0]);  // pt = 1 (LE byte 0)
  for (int r = 1; r < 10; ++r) {
    s0 = _mm_aesenc_si128(s0, k[r]);
    s1 = _mm_aesenc_si128(s1, k[r]);
  }
  s0 = _mm_aesenclast_si128(s0, k[10]);
  s1 = _mm_aesenclast_si128(s1, k[10]);
  _mm_storeu_si128(reinterpret_cast<__m128i*>(out0), s0);
  _mm_storeu_si128(reinterpret_cast<__m128i*>(out1), s1);
}

}  // namespace gpudpf
#else
namespace gpudpf {
bool aesni_available() { return false; }
void aes128_expand_key_ni(const unsigned char[16], AesNiRoundKeys&) {}
void aes128_encrypt2_ni(const AesNiRoundKeys&, unsigned char[16],
                        unsigned char[16]) {}
}  // namespace gpudpf
#endif
