// DPF key generation / evaluation / serialization / layout (see dpf_core.h).

#include "dpf_core.h"

#include <vector>

namespace gpudpf {

// ---------------------------------------------------------------------------
// RNG
// ---------------------------------------------------------------------------
KeyRng::KeyRng(const unsigned char* seed, size_t len) {
  // Absorb the seed into a 128-bit chaining value with Davies-Meyer over
  // AES-128 (h <- E_m(h) ^ h per 16-byte block, bit length appended), then
  // expand the digest as the AES-128-CTR key.
  unsigned char h[16] = {};
  unsigned char blk[16];
  auto absorb = [&h](const unsigned char m[16]) {
    unsigned char e[16];
    aes128_encrypt_block(m, h, e);
    for (int i = 0; i < 16; ++i) h[i] ^= e[i];
  };
  for (size_t off = 0; off < len; off += 16) {
    std::memset(blk, 0, sizeof(blk));
    const size_t take = len - off < 16 ? len - off : 16;
    std::memcpy(blk, seed + off, take);
    absorb(blk);
  }
  std::memset(blk, 0, sizeof(blk));
  const u64 bits = (u64)len * 8;
  std::memcpy(blk, &bits, sizeof(bits));
  absorb(blk);
  aes128_expand_key(h, rk_);
}

u128 KeyRng::next_u128() {
  unsigned char in[16], out[16];
  const u128 c = ctr_++;
  std::memcpy(in, &c, sizeof(in));
  aes128_encrypt_block_rk(rk_, in, out);
  u128 r;
  std::memcpy(&r, out, sizeof(r));
  return r;
}

u128 KeyRng::next_odd_u128() { return next_u128() | 1; }

// ---------------------------------------------------------------------------
// Key generation (iterative; O(log n) PRF calls)
// ---------------------------------------------------------------------------
// Eval recurrence (per key k, index idx):
//   key = root
//   for i = depth-1 .. 0:  b = idx & 1
//     key = PRF(key, b) + cw[key & 1][i*2 + b]   (mod 2^128)
//     idx >>= 1
// Level i consumes index bit (depth-1-i).  The generator walks the on-path
// seeds (s0, s1) top-down, maintaining the invariant that after the level
// using payload beta_i we have s0 - s1 == beta_i (odd => parities differ,
// so the two servers always select opposite correction words on the path;
// off the path the seeds coincide and all contributions cancel).
void dpf_gen(u64 alpha, u128 beta, u64 n, int prf_method, KeyRng& rng,
             DpfKey& k0, DpfKey& k1) {
  if (n < 2 || (n & (n - 1)) != 0)
    throw std::invalid_argument("n must be a power of two >= 2");
  if (alpha >= n) throw std::invalid_argument("alpha must be < n");
  const int depth = ilog2_u64(n);
  if (depth > kMaxDepth) throw std::invalid_argument("n too large");

  k0 = DpfKey{};
  k1 = DpfKey{};
  k0.depth = k1.depth = depth;
  k0.n = k1.n = n;

  // Per-level payloads: the innermost level (i = depth-1, consumed first)
  // through i = 1 carry fresh random odd payloads; the outermost level
  // (i = 0, consumed last) carries the caller's beta.
  std::vector<u128> betas((size_t)depth);
  betas[0] = beta;
  for (int i = 1; i < depth; ++i) betas[(size_t)i] = rng.next_odd_u128();

  // Root seeds with forced opposite parity (server 0 even, server 1 odd).
  u128 s0 = rng.next_u128() & ~(u128)1;
  u128 s1 = rng.next_u128() | 1;
  k0.root = s0;
  k1.root = s1;

  for (int i = depth - 1; i >= 0; --i) {
    const int level_bit = (int)((alpha >> (depth - 1 - i)) & 1);
    const u128 beta_i = betas[(size_t)i];

    // Correction word pair for this level.  Server with even seed adds
    // cw[0], odd seed adds cw[1]; on the path parities differ, so the pair
    // difference must cancel the PRF divergence and inject beta_i at the
    // target bit.  Writing e = the even-seed server's value, o = the odd
    // one's:  e_val - o_val == [bit == level_bit] * sgn * beta_i where sgn
    // accounts for which server holds the even seed.
    u128 cw_path[2];  // the codeword pair at b == level_bit (for the advance)
    for (int b = 0; b < 2; ++b) {
      u128 p0 = prf_eval(prf_method, s0, (u128)b);
      u128 p1 = prf_eval(prf_method, s1, (u128)b);
      u128 rnd = rng.next_u128();
      // Let cw[sel(s0)] = rnd; solve cw[sel(s1)] so that
      //   (p0 + cw[sel0]) - (p1 + cw[sel1]) == (b == level_bit) ? beta_i : 0
      u128 target = (b == level_bit) ? beta_i : (u128)0;
      u128 other = p0 + rnd - p1 - target;  // cw[sel1]
      int sel0 = (int)(s0 & 1);
      u128 cw_pair[2];
      cw_pair[sel0] = rnd;
      cw_pair[sel0 ^ 1] = other;
      k0.cw[0][i * 2 + b] = k1.cw[0][i * 2 + b] = cw_pair[0];
      k0.cw[1][i * 2 + b] = k1.cw[1][i * 2 + b] = cw_pair[1];
      if (b == level_bit) {
        cw_path[0] = cw_pair[0];
        cw_path[1] = cw_pair[1];
      }
    }

    // Advance the on-path seeds through this level.
    u128 n0 = prf_eval(prf_method, s0, (u128)level_bit) +
              cw_path[(size_t)(s0 & 1)];
    u128 n1 = prf_eval(prf_method, s1, (u128)level_bit) +
              cw_path[(size_t)(s1 & 1)];
    s0 = n0;
    s1 = n1;
    if ((u128)(s0 - s1) != beta_i)
      throw std::logic_error("dpf_gen: level invariant violated");
    if (i > 0 && ((s0 ^ s1) & 1) == 0)
      throw std::logic_error("dpf_gen: parity invariant violated");
  }
}

// ---------------------------------------------------------------------------
// Evaluation
// ---------------------------------------------------------------------------
u128 dpf_eval_point(const DpfKey& k, u64 idx, int prf_method) {
  u128 key = k.root;
  u64 rem = idx;
  for (int i = k.depth - 1; i >= 0; --i) {
    int b = (int)(rem & 1);
    u128 v = prf_eval(prf_method, key, (u128)b);
    key = v + k.cw[(size_t)(key & 1)][i * 2 + b];
    rem >>= 1;
  }
  return key;
}

namespace {

// Expand both children of `seed` at eval-level i.  For AES, share one key
// schedule across the two encryptions (the reference re-expands per call;
// see SURVEY.md "hard parts" item 3).
inline void expand_children(const DpfKey& k, int prf_method, u128 seed, int i,
                            u128& c0, u128& c1) {
  u128 p0, p1;
  if (prf_method == PRF_AES128 && aesni_available()) {
    unsigned char key[16], o0[16], o1[16];
    std::memcpy(key, &seed, 16);
    AesNiRoundKeys rk;
    aes128_expand_key_ni(key, rk);
    aes128_encrypt2_ni(rk, o0, o1);
    std::memcpy(&p0, o0, 16);
    std::memcpy(&p1, o1, 16);
  } else if (prf_method == PRF_AES128) {
    unsigned char key[16], in0[16] = {0}, in1[16] = {0}, out[16];
    u32 rk[44];
    std::memcpy(key, &seed, 16);
    aes128_expand_key(key, rk);
    in1[0] = 1;
    aes128_encrypt_block_rk(rk, in0, out);
    std::memcpy(&p0, out, 16);
    aes128_encrypt_block_rk(rk, in1, out);
    std::memcpy(&p1, out, 16);
  } else {
    p0 = prf_eval(prf_method, seed, 0);
    p1 = prf_eval(prf_method, seed, 1);
  }
  int sel = (int)(seed & 1);
  c0 = p0 + k.cw[sel][i * 2 + 0];
  c1 = p1 + k.cw[sel][i * 2 + 1];
}

}  // namespace

// Full-domain expansion, natural order, O(n) PRF pairs (the reference's CPU
// expansion is O(n log n) single calls: dpf_wrapper.cu:70-84).  Node arrays
// are indexed by the partial natural index (consumed bit ell at weight
// 2^ell), so leaves land in natural order directly.
void dpf_expand_full(const DpfKey& k, int prf_method, u32* out) {
  const int depth = k.depth;
  const bool vec8 = avx2_available() &&
                    (prf_method == PRF_SALSA20 || prf_method == PRF_CHACHA20);
  auto x8 = (prf_method == PRF_SALSA20) ? salsa12_x8 : chacha12_x8;
  std::vector<u128> cur(1, k.root), next;
  u128 p0v[8], p1v[8];
  for (int l = 0; l < depth; ++l) {
    const int i = depth - 1 - l;  // eval-order level index
    const u64 width = (u64)1 << l;
    const u64 w8 = vec8 ? (width & ~(u64)7) : 0;
    if (i != 0) next.resize(width * 2);
    // vectorized body: 8 parents per call, one call per child position
    for (u64 v = 0; v < w8; v += 8) {
      x8(&cur[v], 0, p0v);
      x8(&cur[v], 1, p1v);
      for (int j = 0; j < 8; ++j) {
        const int sel = (int)(cur[v + j] & 1);
        u128 c0 = p0v[j] + k.cw[sel][i * 2 + 0];
        u128 c1 = p1v[j] + k.cw[sel][i * 2 + 1];
        if (i == 0) {
          out[v + j] = (u32)c0;
          out[(v + j) | ((u64)1 << l)] = (u32)c1;
        } else {
          next[v + j] = c0;
          next[(v + j) | ((u64)1 << l)] = c1;
        }
      }
    }
    for (u64 v = w8; v < width; ++v) {
      u128 c0, c1;
      expand_children(k, prf_method, cur[v], i, c0, c1);
      if (i == 0) {
        out[v] = (u32)c0;
        out[v | ((u64)1 << l)] = (u32)c1;
      } else {
        next[v] = c0;
        next[v | ((u64)1 << l)] = c1;
      }
    }
    if (i != 0) cur.swap(next);
  }
  if (depth == 0) out[0] = (u32)k.root;
}

void dpf_eval_fused_cpu(const DpfKey& k, int prf_method, const u32* table,
                        int entry_words, u32* out) {
  std::vector<u32> shares(k.n);
  dpf_expand_full(k, prf_method, shares.data());
  for (int m = 0; m < entry_words; ++m) out[m] = 0;
  for (u64 idx = 0; idx < k.n; ++idx) {
    u32 s = shares[idx];
    const u32* row = table + idx * (u64)entry_words;
    for (int m = 0; m < entry_words; ++m) out[m] += s * row[m];
  }
}

// ---------------------------------------------------------------------------
// Serialization (wire format parity with the reference; dpf_wrapper.cu:26-46)
// ---------------------------------------------------------------------------
void key_serialize(const DpfKey& k, std::int32_t out[kKeyInts]) {
  std::memset(out, 0, sizeof(std::int32_t) * kKeyInts);
  u128* slots = reinterpret_cast<u128*>(out);
  slots[0] = (u128)(unsigned)k.depth;
  for (int i = 0; i < 64; ++i) {
    slots[1 + i] = k.cw[0][i];
    slots[65 + i] = k.cw[1][i];
  }
  slots[129] = k.root;
  slots[130] = (u128)k.n;
}

void key_deserialize(const std::int32_t in[kKeyInts], DpfKey& k) {
  const u128* slots = reinterpret_cast<const u128*>(in);
  k.depth = (int)(u64)slots[0];
  if (k.depth < 1 || k.depth > kMaxDepth)
    throw std::invalid_argument("corrupt key: bad depth");
  for (int i = 0; i < 64; ++i) {
    k.cw[0][i] = slots[1 + i];
    k.cw[1][i] = slots[65 + i];
  }
  k.root = slots[129];
  k.n = (u64)slots[130];
  if (k.n != ((u64)1 << k.depth))
    throw std::invalid_argument("corrupt key: n != 2^depth");
}

// ---------------------------------------------------------------------------
// sqrt(n) grid construction
// ---------------------------------------------------------------------------
void grid_dpf_gen(u64 alpha, u128 beta, u64 n_keys, u64 n_codewords,
                  int prf_method, KeyRng& rng, GridDpfKey& k0, GridDpfKey& k1) {
  const u64 n = n_keys * n_codewords;
  if (alpha >= n) throw std::invalid_argument("alpha must be < n");
  const u64 jt = alpha % n_keys;       // target column
  const u64 it = alpha / n_keys;       // target row
  k0 = GridDpfKey{}; k1 = GridDpfKey{};
  k0.n_keys = k1.n_keys = n_keys;
  k0.n_codewords = k1.n_codewords = n_codewords;
  k0.seeds.resize(n_keys); k1.seeds.resize(n_keys);
  for (int s = 0; s < 2; ++s) { k0.cw[s].resize(n_codewords); k1.cw[s].resize(n_codewords); }

  for (u64 j = 0; j < n_keys; ++j) {
    if (j == jt) {
      k0.seeds[j] = rng.next_u128() & ~(u128)1;  // forced even
      k1.seeds[j] = rng.next_u128() | 1;         // forced odd
    } else {
      k0.seeds[j] = k1.seeds[j] = rng.next_u128();
    }
  }
  const u128 s0 = k0.seeds[jt], s1 = k1.seeds[jt];
  const int sel0 = (int)(s0 & 1);
  for (u64 i = 0; i < n_codewords; ++i) {
    u128 p0 = prf_eval(prf_method, s0, (u128)i);
    u128 p1 = prf_eval(prf_method, s1, (u128)i);
    u128 rnd = rng.next_u128();
    u128 target = (i == it) ? beta : (u128)0;
    u128 other = p0 + rnd - p1 - target;
    k0.cw[sel0][i] = k1.cw[sel0][i] = rnd;
    k0.cw[sel0 ^ 1][i] = k1.cw[sel0 ^ 1][i] = other;
  }
}

u128 grid_dpf_eval(const GridDpfKey& k, u64 idx, int prf_method) {
  const u128 seed = k.seeds[idx % k.n_keys];
  const u64 row = idx / k.n_keys;
  return prf_eval(prf_method, seed, (u128)row) + k.cw[(size_t)(seed & 1)][row];
}

// ---------------------------------------------------------------------------
// Layout contract
// ---------------------------------------------------------------------------
int zlog_for_depth(int depth) {
  // Z = workgroup size of the fused kernel (256 threads = 4 wave64), capped
  // so each thread keeps a subtree of at least one leaf pair (DS >= 1).
  int zlog = 8;
  if (depth - 1 < zlog) zlog = depth - 1;
  if (zlog < 0) zlog = 0;
  return zlog;
}

namespace {
inline u64 bitrev(u64 v, int bits) {
  u64 r = 0;
  for (int i = 0; i < bits; ++i) r |= ((v >> i) & 1) << (bits - 1 - i);
  return r;
}
}  // namespace

u64 leaf_perm(u64 n, int zlog, u64 idx) {
  const int depth = ilog2_u64(n);
  const int ds = depth - zlog;  // subtree splits per thread (>= 1)
  const u64 t = bitrev(idx & ((n > 1 ? ((u64)1 << zlog) : 1) - 1), zlog);
  const u64 mid = (ds >= 2) ? ((idx >> zlog) & (((u64)1 << (ds - 1)) - 1)) : 0;
  const u64 j = bitrev(mid, ds - 1);
  const u64 b = (idx >> (depth - 1)) & 1;
  return (j << (zlog + 1)) | (t << 1) | b;
}

u64 leaf_perm_inv(u64 n, int zlog, u64 row) {
  const int depth = ilog2_u64(n);
  const int ds = depth - zlog;
  const u64 b = row & 1;
  const u64 t = (row >> 1) & (((u64)1 << zlog) - 1);
  const u64 j = row >> (zlog + 1);
  u64 idx = bitrev(t, zlog);
  idx |= bitrev(j, ds - 1) << zlog;
  idx |= b << (depth - 1);
  return idx;
}

// ---------------------------------------------------------------------------
// Sharding
// ---------------------------------------------------------------------------
void dpf_shard_subkey(const DpfKey& k, int prf_method, u64 rank, u64 world,
                      DpfKey& out) {
  if (world < 1 || (world & (world - 1)) != 0)
    throw std::invalid_argument("world size must be a power of two");
  if (rank >= world) throw std::invalid_argument("rank must be < world");
  const int wlog = ilog2_u64(world);
  if (wlog >= k.depth)
    throw std::invalid_argument("world size too large for key depth");

  out = DpfKey{};
  out.depth = k.depth - wlog;
  out.n = k.n >> wlog;
  // Walk wlog levels from the root consuming rank bits LSB-first; the
  // remaining correction words (eval levels 0..depth-wlog-1) carry over
  // unchanged because eval level indices count from the leaf end.
  u128 key = k.root;
  for (int l = 0; l < wlog; ++l) {
    const int i = k.depth - 1 - l;
    const int b = (int)((rank >> l) & 1);
    u128 v = prf_eval(prf_method, key, (u128)b);
    key = v + k.cw[(size_t)(key & 1)][i * 2 + b];
  }
  out.root = key;
  for (int i = 0; i < out.depth; ++i) {
    for (int b = 0; b < 2; ++b) {
      out.cw[0][i * 2 + b] = k.cw[0][i * 2 + b];
      out.cw[1][i * 2 + b] = k.cw[1][i * 2 + b];
    }
  }
}

}  // namespace gpudpf
