// gpudpf CPU core: distributed point function (DPF) key generation and
// reference evaluation for 2-server PIR, plus the table-layout contract
// shared with the MI355X HIP kernels.
//
// Capability parity with facebookresearch/GPU-DPF:
//   - log(n) GGM-tree DPF with per-level parity-selected correction words
//     (reference: dpf_base/dpf.h:403-464, DPF.md:45-91)
//   - wire format: 524 x int32 = 131 u128 slots
//     [depth][cw_1[64]][cw_2[64]][root][n]  (reference: dpf_wrapper.cu:26-46)
//   - PRFs: DUMMY / SALSA20(12) / CHACHA20(12) / AES128, bit-exact with the
//     reference semantics (reference: dpf_base/dpf.h:65-235); verified by
//     golden vectors in tests/test_prf_vectors.py.
//
// The implementation is written MI355X-first and from scratch: key
// generation is ITERATIVE (O(log n) PRF calls instead of the reference's
// O(log^2 n) recursive re-evaluation), and the evaluation layout contract
// (leaf_perm) is designed for the wave64 per-thread-DFS kernel in
// csrc/hip/dpf_kernels.hip rather than the reference's Z-frontier DFS.

#pragma once
#include <cstdint>
#include <cstring>
#include <cstddef>
#include <stdexcept>
#include <vector>

namespace gpudpf {

using u128 = unsigned __int128;
using u64 = std::uint64_t;
using u32 = std::uint32_t;

enum PrfMethod : int {
  PRF_DUMMY = 0,
  PRF_SALSA20 = 1,
  PRF_CHACHA20 = 2,
  PRF_AES128 = 3,
};

// ---------------------------------------------------------------------------
// PRFs (prf.cc / aes128.cc).  Contract: prf(seed: u128, pos: u128) -> u128.
// In the log(n) DPF, pos is only ever 0 or 1.
// ---------------------------------------------------------------------------
u128 prf_dummy(u128 seed, u128 pos);
u128 prf_salsa20_12(u128 seed, u128 pos);
u128 prf_chacha20_12(u128 seed, u128 pos);
u128 prf_aes128(u128 seed, u128 pos);
u128 prf_eval(int method, u128 seed, u128 pos);

// Standard AES-128 ECB single-block encrypt (FIPS-197), used by
// prf_aes128 and exposed for test vectors.
void aes128_encrypt_block(const unsigned char key[16], const unsigned char in[16],
                          unsigned char out[16]);
// Expanded AES-128 key schedule (11 round keys x 4 words, byte-packed
// big-endian-within-word as used by the T-table cipher path).
void aes128_expand_key(const unsigned char key[16], u32 rk[44]);
void aes128_encrypt_block_rk(const u32 rk[44], const unsigned char in[16],
                             unsigned char out[16]);
// Table generators for the GPU path (host computes, device copies to LDS):
// te[0..3][256] round T-tables, sbox[256] (as u32).
void aes128_tables(u32 te0[256], u32 te1[256], u32 te2[256], u32 te3[256],
                   u32 sbox[256]);

// AES-NI fast path (x86 hosts; runtime-guarded, bit-exact with the table
// implementation).  Used by the hot CPU expansion loop.
struct AesNiRoundKeys {
  alignas(16) unsigned char rk[11 * 16];
};
bool aesni_available();
// AVX2 8-lane stream-cipher PRFs (runtime-guarded; prf_avx2.cc)
bool avx2_available();
void salsa12_x8(const u128* seeds, u32 pos, u128* out);
void chacha12_x8(const u128* seeds, u32 pos, u128* out);
void aes128_expand_key_ni(const unsigned char key[16], AesNiRoundKeys& rk);
void aes128_encrypt2_ni(const AesNiRoundKeys& rk, unsigned char out0[16],
                        unsigned char out1[16]);

// ---------------------------------------------------------------------------
// DPF key
// ---------------------------------------------------------------------------
// The 524-int wire format carries 64 correction-word slots per selector =
// 2*depth entries, so depth <= 32 (n <= 2^32) is a hard wire-format limit:
// dpf_gen and key_deserialize both enforce it (a deeper key would serialize
// silently truncated).
constexpr int kMaxDepth = 32;
constexpr int kKeyInts = 524;   // serialized size in int32 (2096 bytes)
constexpr int kEntryWords = 16; // table entry = 16 x u32 (padded)

struct DpfKey {
  int depth = 0;          // log2(n)
  u64 n = 0;              // table size
  u128 root = 0;          // this server's root seed
  u128 cw[2][2 * kMaxDepth] = {}; // cw[sel][level*2 + bit]; level in eval order
};

// Seedable RNG for key generation (deterministic given seed bytes).
// AES-128-CTR keyed by a Davies-Meyer digest of the seed bytes: unlike a
// Mersenne Twister (which the reference uses, dpf_base/dpf.h RandGen —
// and whose raw outputs, published verbatim as correction words, let one
// server linearly reconstruct the generator state and recover the other
// server's seeds), the published cw stream reveals nothing about the
// generator key.  Callers pass os.urandom entropy for production keygen;
// fixed seeds give deterministic keys for tests.
class KeyRng {
 public:
  explicit KeyRng(const unsigned char* seed, size_t len);
  u128 next_u128();
  u128 next_odd_u128();
 private:
  u32 rk_[44];   // AES-128 round keys (CTR mode)
  u128 ctr_ = 0;
};

// Generate the two server keys for point alpha with payload beta over a
// domain of n entries (n a power of two, n >= 2).  Iterative construction:
// walk the on-path seeds from the root down, emitting one correction-word
// pair per level.  Semantics match the reference scheme exactly (see file
// header); asserts the per-level invariant s0 - s1 == beta_level.
void dpf_gen(u64 alpha, u128 beta, u64 n, int prf_method, KeyRng& rng,
             DpfKey& k0, DpfKey& k1);

// Evaluate one index: O(depth) PRF calls.  This is the correctness anchor
// the GPU kernels are tested against (reference: EvaluateFlat semantics,
// dpf_base/dpf.h:362-377).
u128 dpf_eval_point(const DpfKey& k, u64 idx, int prf_method);

// Expand the full domain into low-32-bit shares in NATURAL index order.
// (Output truncation to u32 is exact: mod 2^32 is a ring hom of mod 2^128.)
void dpf_expand_full(const DpfKey& k, int prf_method, u32* out);

// Fused expand + table inner product on CPU (reference oracle for the GPU
// fused kernel): out[m] = sum_i share_i * table[i*stride+m]  (mod 2^32),
// table in NATURAL order.
void dpf_eval_fused_cpu(const DpfKey& k, int prf_method, const u32* table,
                        int entry_words, u32* out);

// Serialization: 524 int32 (see file header for slot map).
void key_serialize(const DpfKey& k, std::int32_t out[kKeyInts]);
void key_deserialize(const std::int32_t in[kKeyInts], DpfKey& k);

// ---------------------------------------------------------------------------
// Layout contract shared with the HIP kernels
// ---------------------------------------------------------------------------
// The fused kernel runs one key per workgroup of Z = 1<<zlog threads.
// Phase 1 expands the GGM tree breadth-first for zlog levels (frontier
// position t = bitrev_zlog(first-consumed index bits)).  Phase 2: thread t
// DFS-expands its subtree emitting leaf PAIRS in DFS order j; the final
// level's bit is b.  The table is stored so that at step j the workgroup
// reads one contiguous slab:
//     row(idx) = j << (zlog+1) | t << 1 | b
// with  t = bitrev_zlog(idx & (Z-1)),
//       j = bitrev_{DS-1}((idx >> zlog) & (2^{DS-1}-1)),  DS = depth - zlog,
//       b = idx >> (depth-1).
int zlog_for_depth(int depth);
u64 leaf_perm(u64 n, int zlog, u64 idx);       // natural -> permuted row
u64 leaf_perm_inv(u64 n, int zlog, u64 row);   // permuted row -> natural

// Multi-GPU row sharding: rank r of W (W a power of two) owns natural
// indices with idx % W == r; the restriction of a DPF key to that residue
// class is itself a DPF key of depth-log2(W) obtained by walking log2(W)
// levels from the root consuming the bits of r LSB-first.
void dpf_shard_subkey(const DpfKey& k, int prf_method, u64 rank, u64 world,
                      DpfKey& out);

// ---------------------------------------------------------------------------
// sqrt(n) GRID construction (the reference's general n_keys x n_codewords
// DPF, dpf_base/dpf.h:290-360): one seed per column, one codeword pair
// per row; eval(idx) = PRF(seed[idx % n_keys], idx / n_keys)
//                      + cw[seed parity][idx / n_keys].
// Kept as a standalone research component (the log-n scheme uses it only
// conceptually as its N=2 base case here).
// ---------------------------------------------------------------------------
struct GridDpfKey {
  u64 n_keys = 0, n_codewords = 0;
  std::vector<u128> seeds;      // [n_keys]
  std::vector<u128> cw[2];      // [n_codewords] each; selected by parity
};

void grid_dpf_gen(u64 alpha, u128 beta, u64 n_keys, u64 n_codewords,
                  int prf_method, KeyRng& rng, GridDpfKey& k0, GridDpfKey& k1);
u128 grid_dpf_eval(const GridDpfKey& k, u64 idx, int prf_method);

inline int ilog2_u64(u64 v) {
  int l = 0;
  while ((u64(1) << l) < v) ++l;
  return l;
}

}  // namespace gpudpf
