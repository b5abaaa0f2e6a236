// CPU PRFs for the DPF.  Contract: PRF(seed: u128, pos: u128) -> u128,
// bit-exact with the reference scheme so that keys interoperate
// (reference: dpf_base/dpf.h:65-235).  The stream-cipher PRFs are the
// standard Salsa20/12 and ChaCha20/12 block functions with the reference's
// specific i/o conventions (which words of the 16-word state carry the
// 128-bit seed / the position, and which output words form the result).
// Golden vectors: tests/test_prf_vectors.py.

#include "dpf_core.h"

namespace gpudpf {

namespace {
inline u32 rotl32(u32 v, int s) { return (v << s) | (v >> (32 - s)); }
}

u128 prf_dummy(u128 seed, u128 pos) {
  return seed * (pos + 4242) + (pos + 4242);
}

// Salsa20/12 block.  Seed occupies state words 1..4 (seed bits 127..96 in
// word 1 down to bits 31..0 in word 4); pos occupies words 8 (bits 63..32)
// and 9 (bits 31..0); the four "expand 32-byte k" constants sit at words
// 0, 5, 10, 15.  Result = output words 1..4 assembled high-to-low.
u128 prf_salsa20_12(u128 seed, u128 pos) {
  u32 in[16] = {0};
  in[1] = (u32)(seed >> 96);
  in[2] = (u32)(seed >> 64);
  in[3] = (u32)(seed >> 32);
  in[4] = (u32)seed;
  in[8] = (u32)(pos >> 32);
  in[9] = (u32)pos;
  in[0] = 0x65787061u;
  in[5] = 0x6e642033u;
  in[10] = 0x322d6279u;
  in[15] = 0x7465206bu;

  u32 x[16];
  for (int i = 0; i < 16; ++i) x[i] = in[i];
  for (int r = 0; r < 6; ++r) {
    // column round
    auto qr = [](u32& a, u32& b, u32& c, u32& d) {
      b ^= rotl32(a + d, 7);
      c ^= rotl32(b + a, 9);
      d ^= rotl32(c + b, 13);
      a ^= rotl32(d + c, 18);
    };
    qr(x[0], x[4], x[8], x[12]);
    qr(x[5], x[9], x[13], x[1]);
    qr(x[10], x[14], x[2], x[6]);
    qr(x[15], x[3], x[7], x[11]);
    // row round
    qr(x[0], x[1], x[2], x[3]);
    qr(x[5], x[6], x[7], x[4]);
    qr(x[10], x[11], x[8], x[9]);
    qr(x[15], x[12], x[13], x[14]);
  }
  return ((u128)(x[1] + in[1]) << 96) | ((u128)(x[2] + in[2]) << 64) |
         ((u128)(x[3] + in[3]) << 32) | (u128)(x[4] + in[4]);
}

// ChaCha20/12 block.  Seed in state words 4..7 (high-to-low), pos in words
// 12 (bits 63..32) and 13 (bits 31..0), constants in words 0..3.
// Result = output words 4..7 assembled high-to-low.
u128 prf_chacha20_12(u128 seed, u128 pos) {
  u32 in[16] = {0};
  in[4] = (u32)(seed >> 96);
  in[5] = (u32)(seed >> 64);
  in[6] = (u32)(seed >> 32);
  in[7] = (u32)seed;
  in[12] = (u32)(pos >> 32);
  in[13] = (u32)pos;
  in[0] = 0x65787061u;
  in[1] = 0x6e642033u;
  in[2] = 0x322d6279u;
  in[3] = 0x7465206bu;

  u32 x[16];
  for (int i = 0; i < 16; ++i) x[i] = in[i];
  auto qr = [](u32& a, u32& b, u32& c, u32& d) {
    a += b; d ^= a; d = rotl32(d, 16);
    c += d; b ^= c; b = rotl32(b, 12);
    a += b; d ^= a; d = rotl32(d, 8);
    c += d; b ^= c; b = rotl32(b, 7);
  };
  for (int r = 0; r < 6; ++r) {
    qr(x[0], x[4], x[8], x[12]);
    qr(x[1], x[5], x[9], x[13]);
    qr(x[2], x[6], x[10], x[14]);
    qr(x[3], x[7], x[11], x[15]);
    qr(x[0], x[5], x[10], x[15]);
    qr(x[1], x[6], x[11], x[12]);
    qr(x[2], x[7], x[8], x[13]);
    qr(x[3], x[4], x[9], x[14]);
  }
  return ((u128)(x[4] + in[4]) << 96) | ((u128)(x[5] + in[5]) << 64) |
         ((u128)(x[6] + in[6]) << 32) | (u128)(x[7] + in[7]);
}

// AES-128: pos (as a 16-byte little-endian block) encrypted under the seed
// (16-byte little-endian key); ciphertext read back little-endian.
u128 prf_aes128(u128 seed, u128 pos) {
  unsigned char key[16], in[16], out[16];
  std::memcpy(key, &seed, 16);
  std::memcpy(in, &pos, 16);
  aes128_encrypt_block(key, in, out);
  u128 r;
  std::memcpy(&r, out, 16);
  return r;
}

u128 prf_eval(int method, u128 seed, u128 pos) {
  switch (method) {
    case PRF_DUMMY: return prf_dummy(seed, pos);
    case PRF_SALSA20: return prf_salsa20_12(seed, pos);
    case PRF_CHACHA20: return prf_chacha20_12(seed, pos);
    case PRF_AES128: return prf_aes128(seed, pos);
    default: throw std::invalid_argument("unknown PRF method");
  }
}

}  // namespace gpudpf
