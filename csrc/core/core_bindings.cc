// pybind11 bindings for the CPU DPF core -> python module gpudpf._core.
// Keys cross the boundary as int32[524] numpy arrays (the torch wrapper in
// gpudpf/dpf.py converts to/from torch tensors zero-copy).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <cstring>
#include <thread>
#include <vector>

#include "dpf_core.h"

namespace py = pybind11;
using namespace gpudpf;

namespace {

using KeyArr = py::array_t<std::int32_t, py::array::c_style>;

KeyArr key_to_array(const DpfKey& k) {
  KeyArr arr(kKeyInts);
  key_serialize(k, arr.mutable_data());
  return arr;
}

DpfKey key_from_array(const KeyArr& arr) {
  if (arr.size() != kKeyInts) throw std::invalid_argument("key must have 524 int32s");
  DpfKey k;
  key_deserialize(arr.data(), k);
  return k;
}

py::tuple gen(u64 alpha, u64 n, py::bytes seed, int prf_method) {
  std::string s = seed;
  KeyRng rng(reinterpret_cast<const unsigned char*>(s.data()), s.size());
  DpfKey k0, k1;
  dpf_gen(alpha, /*beta=*/1, n, prf_method, rng, k0, k1);
  return py::make_tuple(key_to_array(k0), key_to_array(k1));
}

// Batched key generation: alphas int64[B] -> (k0s, k1s) int32[B,524].
// One seeded RNG stream covers the whole batch (deterministic).
py::tuple gen_batch(py::array_t<std::int64_t, py::array::c_style | py::array::forcecast> alphas,
                    u64 n, py::bytes seed, int prf_method) {
  std::string sd = seed;
  KeyRng rng(reinterpret_cast<const unsigned char*>(sd.data()), sd.size());
  const py::ssize_t b = alphas.size();
  py::array_t<std::int32_t> k0s({b, (py::ssize_t)kKeyInts});
  py::array_t<std::int32_t> k1s({b, (py::ssize_t)kKeyInts});
  const std::int64_t* ap = alphas.data();
  std::int32_t* p0 = k0s.mutable_data();
  std::int32_t* p1 = k1s.mutable_data();
  {
    py::gil_scoped_release nogil;
    DpfKey k0, k1;
    for (py::ssize_t i = 0; i < b; ++i) {
      dpf_gen((u64)ap[i], 1, n, prf_method, rng, k0, k1);
      key_serialize(k0, p0 + i * kKeyInts);
      key_serialize(k1, p1 + i * kKeyInts);
    }
  }
  return py::make_tuple(k0s, k1s);
}

// Full-domain expansion to low-32 shares, natural order -> int32[n].
py::array_t<std::int32_t> expand(const KeyArr& key, int prf_method) {
  DpfKey k = key_from_array(key);
  py::array_t<std::int32_t> out((py::ssize_t)k.n);
  {
    py::gil_scoped_release nogil;
    dpf_expand_full(k, prf_method, reinterpret_cast<u32*>(out.mutable_data()));
  }
  return out;
}

// Batched expansion with a thread pool (CPU serving path; the reference's
// CPU baseline is OpenMP-parallel google-DPF expansion,
// paper/kernel/cpu/dpf_google/benchmark.cu:117-120).
py::array_t<std::int32_t> expand_batch(const std::vector<KeyArr>& keys,
                                       int prf_method, int num_threads) {
  const size_t b = keys.size();
  if (b == 0) return py::array_t<std::int32_t>(0);
  std::vector<DpfKey> ks(b);
  for (size_t i = 0; i < b; ++i) ks[i] = key_from_array(keys[i]);
  const u64 n = ks[0].n;
  for (auto& k : ks)
    if (k.n != n) throw std::invalid_argument("keys must share a domain size");
  py::array_t<std::int32_t> out({(py::ssize_t)b, (py::ssize_t)n});
  u32* optr = reinterpret_cast<u32*>(out.mutable_data());
  {
    py::gil_scoped_release nogil;
    if (num_threads <= 1) {
      for (size_t i = 0; i < b; ++i)
        dpf_expand_full(ks[i], prf_method, optr + i * n);
    } else {
      // When keys are scarcer than threads, split each key's expansion
      // into W subtree shards (the residue-class restriction is itself a
      // DPF key; shard r's leaves are the strided slice out[r::W]).
      u64 shards = 1;
      while (b * shards < (size_t)num_threads &&
             shards * 2 <= (ks[0].n >> 1) && shards < 64)
        shards <<= 1;
      std::vector<std::thread> pool;
      std::atomic<size_t> next{0};
      const size_t tasks = b * (size_t)shards;
      for (int t = 0; t < num_threads; ++t) {
        pool.emplace_back([&, shards] {
          std::vector<u32> tmp(shards > 1 ? ks[0].n / shards : 0);
          for (size_t task; (task = next.fetch_add(1)) < tasks;) {
            const size_t i = task / shards;
            const u64 r = (u64)(task % shards);
            if (shards == 1) {
              dpf_expand_full(ks[i], prf_method, optr + i * n);
            } else {
              DpfKey sub;
              dpf_shard_subkey(ks[i], prf_method, r, shards, sub);
              dpf_expand_full(sub, prf_method, tmp.data());
              u32* row = optr + i * n;
              for (u64 j = 0; j < sub.n; ++j) row[j * shards + r] = tmp[j];
            }
          }
        });
      }
      for (auto& th : pool) th.join();
    }
  }
  return out;
}

u64 eval_point_low(const KeyArr& key, u64 idx, int prf_method) {
  DpfKey k = key_from_array(key);
  return (u64)(u32)dpf_eval_point(k, idx, prf_method);
}

py::tuple eval_point128(const KeyArr& key, u64 idx, int prf_method) {
  DpfKey k = key_from_array(key);
  u128 v = dpf_eval_point(k, idx, prf_method);
  return py::make_tuple((u64)v, (u64)(v >> 64));
}

py::array_t<std::int32_t> eval_fused_cpu(const KeyArr& key,
                                         py::array_t<std::int32_t, py::array::c_style | py::array::forcecast> table,
                                         int prf_method) {
  DpfKey k = key_from_array(key);
  if (table.ndim() != 2 || (u64)table.shape(0) != k.n)
    throw std::invalid_argument("table must be [n, entry_words]");
  int ew = (int)table.shape(1);
  py::array_t<std::int32_t> out(ew);
  {
    py::gil_scoped_release nogil;
    dpf_eval_fused_cpu(k, prf_method, reinterpret_cast<const u32*>(table.data()),
                       ew, reinterpret_cast<u32*>(out.mutable_data()));
  }
  return out;
}

// natural -> permuted row map as an int64[n] array (used by eval_init to
// reorder the table, and its inverse by the one-hot output path).
py::array_t<std::int64_t> leaf_perm_table(u64 n, int zlog) {
  py::array_t<std::int64_t> out((py::ssize_t)n);
  auto* p = out.mutable_data();
  {
    py::gil_scoped_release nogil;
    for (u64 i = 0; i < n; ++i) p[i] = (std::int64_t)leaf_perm(n, zlog, i);
  }
  return out;
}

// Permuted rows for a SUBSET of natural indices (leaf_perm is a pure bit
// permutation, so it never needs the full n-entry map): lets eval_init
// stream multi-hundred-GB tables chunk by chunk without materializing an
// 8-byte-per-row permutation table (34 GB at n=2^32).
py::array_t<std::int64_t> leaf_perm_rows(
    py::array_t<std::int64_t, py::array::c_style | py::array::forcecast> idx,
    u64 n, int zlog) {
  const auto m = idx.size();
  py::array_t<std::int64_t> out(m);
  auto* p = out.mutable_data();
  const auto* q = idx.data();
  {
    py::gil_scoped_release nogil;
    for (py::ssize_t i = 0; i < m; ++i) {
      const u64 v = (u64)q[i];
      if (v >= n) throw std::out_of_range("index out of range");
      p[i] = (std::int64_t)leaf_perm(n, zlog, v);
    }
  }
  return out;
}

// Compact wire format: the 524-int format always carries 64 cw slots; a
// depth-d key only uses 2d of them, so shallow-tree keys ship dead
// bytes (the reference's SizeOf, dpf_base/dpf.h:474, counts the same
// padding).  Layout: [depth u128][cw0[0..2d) u128][cw1[0..2d) u128]
// [root u128][n u128] = (3 + 4d) * 16 bytes; depth 14 -> 944 B vs 2096.
// Lossless round-trip with the standard format; same scheme, same keys.
py::array_t<std::int32_t> key_compact(const KeyArr& key) {
  DpfKey k = key_from_array(key);
  const int d = k.depth;
  py::array_t<std::int32_t> out((py::ssize_t)((3 + 4 * d) * 4));
  std::int32_t* op = out.mutable_data();
  u128* slots = reinterpret_cast<u128*>(op);
  slots[0] = (u128)(unsigned)d;
  for (int i = 0; i < 2 * d; ++i) {
    slots[1 + i] = k.cw[0][i];
    slots[1 + 2 * d + i] = k.cw[1][i];
  }
  slots[1 + 4 * d] = k.root;
  slots[2 + 4 * d] = (u128)k.n;
  return out;
}

KeyArr key_expand_compact(
    py::array_t<std::int32_t, py::array::c_style | py::array::forcecast> c) {
  if (c.size() < 8 || c.size() % 4 != 0)
    throw std::invalid_argument("corrupt compact key: bad length");
  const u128* slots = reinterpret_cast<const u128*>(c.data());
  const int d = (int)(u64)slots[0];
  if (d < 1 || d > kMaxDepth)
    throw std::invalid_argument("corrupt compact key: bad depth");
  if (c.size() != (py::ssize_t)((3 + 4 * d) * 4))
    throw std::invalid_argument("corrupt compact key: length != 16*(3+4d)");
  DpfKey k;
  k.depth = d;
  for (int i = 0; i < 2 * d; ++i) {
    k.cw[0][i] = slots[1 + i];
    k.cw[1][i] = slots[1 + 2 * d + i];
  }
  k.root = slots[1 + 4 * d];
  k.n = (u64)slots[2 + 4 * d];
  if (k.n != ((u64)1 << d))
    throw std::invalid_argument("corrupt compact key: n != 2^depth");
  return key_to_array(k);
}

py::tuple prf(int method, u64 seed_lo, u64 seed_hi, u64 pos) {
  u128 s = ((u128)seed_hi << 64) | seed_lo;
  u128 r = prf_eval(method, s, (u128)pos);
  return py::make_tuple((u64)r, (u64)(r >> 64));
}

py::bytes aes_block(py::bytes key, py::bytes block) {
  std::string k = key, b = block;
  if (k.size() != 16 || b.size() != 16)
    throw std::invalid_argument("key and block must be 16 bytes");
  unsigned char out[16];
  aes128_encrypt_block(reinterpret_cast<const unsigned char*>(k.data()),
                       reinterpret_cast<const unsigned char*>(b.data()), out);
  return py::bytes(reinterpret_cast<char*>(out), 16);
}

KeyArr shard_subkey(const KeyArr& key, int prf_method, u64 rank, u64 world) {
  DpfKey k = key_from_array(key);
  DpfKey sub;
  dpf_shard_subkey(k, prf_method, rank, world, sub);
  return key_to_array(sub);
}

// Batched subkey restriction for the distributed hot path: [B,524] int32
// keys -> [B,524] subkeys in one call (one GIL release, no per-key
// python round-trips — the per-key variant costs ~10 us of python
// overhead each, which at batch 512 would dwarf the sharded kernel).
py::array_t<std::int32_t> shard_subkey_batch(
    py::array_t<std::int32_t, py::array::c_style | py::array::forcecast> keys,
    int prf_method, u64 rank, u64 world) {
  if (keys.ndim() != 2 || keys.shape(1) != kKeyInts)
    throw std::invalid_argument("keys must be [B, 524] int32");
  const py::ssize_t b = keys.shape(0);
  py::array_t<std::int32_t> out({b, (py::ssize_t)kKeyInts});
  const std::int32_t* ip = keys.data();
  std::int32_t* op = out.mutable_data();
  {
    py::gil_scoped_release nogil;
    DpfKey k, sub;
    for (py::ssize_t i = 0; i < b; ++i) {
      key_deserialize(ip + i * kKeyInts, k);
      dpf_shard_subkey(k, prf_method, rank, world, sub);
      key_serialize(sub, op + i * kKeyInts);
    }
  }
  return out;
}

// Exact u128 GEMM CPU reference: a [M,K,4] int32 (u128 limbs LE),
// bt [N,K,4] -> c [M,N,4].
py::array_t<std::int32_t> gemm128_cpu(
    py::array_t<std::int32_t, py::array::c_style | py::array::forcecast> a,
    py::array_t<std::int32_t, py::array::c_style | py::array::forcecast> bt) {
  if (a.ndim() != 3 || bt.ndim() != 3 || a.shape(2) != 4 || bt.shape(2) != 4 ||
      a.shape(1) != bt.shape(1))
    throw std::invalid_argument("shapes must be [M,K,4] and [N,K,4]");
  const py::ssize_t M = a.shape(0), N = bt.shape(0), K = a.shape(1);
  py::array_t<std::int32_t> c({M, N, (py::ssize_t)4});
  const u128* ap = reinterpret_cast<const u128*>(a.data());
  const u128* bp = reinterpret_cast<const u128*>(bt.data());
  u128* cp = reinterpret_cast<u128*>(c.mutable_data());
  {
    py::gil_scoped_release nogil;
    for (py::ssize_t m = 0; m < M; ++m)
      for (py::ssize_t n = 0; n < N; ++n) {
        u128 acc = 0;
        for (py::ssize_t k = 0; k < K; ++k) acc += ap[m * K + k] * bp[n * K + k];
        cp[m * N + n] = acc;
      }
  }
  return c;
}

// sqrt(n) grid DPF: gen -> two dicts of numpy arrays; full-domain eval.
py::tuple grid_gen(u64 alpha, u64 n_keys, u64 n_codewords, py::bytes seed,
                   int prf_method) {
  std::string sd = seed;
  KeyRng rng(reinterpret_cast<const unsigned char*>(sd.data()), sd.size());
  GridDpfKey k0, k1;
  grid_dpf_gen(alpha, /*beta=*/1, n_keys, n_codewords, prf_method, rng, k0, k1);
  auto pack = [](const GridDpfKey& k) {
    py::dict d;
    py::array_t<std::int32_t> seeds({(py::ssize_t)k.n_keys, (py::ssize_t)4});
    std::memcpy(seeds.mutable_data(), k.seeds.data(), k.n_keys * 16);
    d["seeds"] = seeds;
    for (int s = 0; s < 2; ++s) {
      py::array_t<std::int32_t> cw({(py::ssize_t)k.n_codewords, (py::ssize_t)4});
      std::memcpy(cw.mutable_data(), k.cw[s].data(), k.n_codewords * 16);
      d[s == 0 ? "cw_even" : "cw_odd"] = cw;
    }
    d["n_keys"] = (long long)k.n_keys;
    d["n_codewords"] = (long long)k.n_codewords;
    return d;
  };
  return py::make_tuple(pack(k0), pack(k1));
}

py::array_t<std::int32_t> grid_expand(py::dict key, int prf_method) {
  GridDpfKey k;
  k.n_keys = (u64)py::cast<long long>(key["n_keys"]);
  k.n_codewords = (u64)py::cast<long long>(key["n_codewords"]);
  auto seeds = py::cast<py::array_t<std::int32_t>>(key["seeds"]);
  auto cwe = py::cast<py::array_t<std::int32_t>>(key["cw_even"]);
  auto cwo = py::cast<py::array_t<std::int32_t>>(key["cw_odd"]);
  k.seeds.resize(k.n_keys);
  k.cw[0].resize(k.n_codewords);
  k.cw[1].resize(k.n_codewords);
  std::memcpy(k.seeds.data(), seeds.data(), k.n_keys * 16);
  std::memcpy(k.cw[0].data(), cwe.data(), k.n_codewords * 16);
  std::memcpy(k.cw[1].data(), cwo.data(), k.n_codewords * 16);
  const u64 n = k.n_keys * k.n_codewords;
  py::array_t<std::int32_t> out((py::ssize_t)n);
  u32* optr = reinterpret_cast<u32*>(out.mutable_data());
  {
    py::gil_scoped_release nogil;
    for (u64 i = 0; i < n; ++i)
      optr[i] = (u32)grid_dpf_eval(k, i, prf_method);
  }
  return out;
}

// AES GPU tables (5 x 256 u32: te0..te3, sbox) for upload to the device.
py::array_t<std::int32_t> aes_gpu_tables() {
  py::array_t<std::int32_t> out(5 * 256);
  u32* p = reinterpret_cast<u32*>(out.mutable_data());
  aes128_tables(p, p + 256, p + 512, p + 768, p + 1024);
  return out;
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "gpudpf CPU core (keygen, reference eval, layout)";
  m.def("gen", &gen, py::arg("alpha"), py::arg("n"), py::arg("seed"),
        py::arg("prf_method"));
  m.def("gen_batch", &gen_batch, py::arg("alphas"), py::arg("n"),
        py::arg("seed"), py::arg("prf_method"));
  m.def("expand", &expand, py::arg("key"), py::arg("prf_method"));
  m.def("expand_batch", &expand_batch, py::arg("keys"), py::arg("prf_method"),
        py::arg("num_threads") = 1);
  m.def("eval_point_low", &eval_point_low);
  m.def("eval_point128", &eval_point128);
  m.def("eval_fused_cpu", &eval_fused_cpu);
  m.def("leaf_perm_table", &leaf_perm_table);
  m.def("leaf_perm_rows", &leaf_perm_rows);
  m.def("zlog_for_depth", &zlog_for_depth);
  m.def("prf", &prf);
  m.def("aes_block", &aes_block);
  m.def("shard_subkey", &shard_subkey);
  m.def("shard_subkey_batch", &shard_subkey_batch);
  m.def("key_compact", &key_compact);
  m.def("key_expand_compact", &key_expand_compact);
  m.def("aes_gpu_tables", &aes_gpu_tables);
  m.def("gemm128_cpu", &gemm128_cpu);
  m.def("grid_gen", &grid_gen, py::arg("alpha"), py::arg("n_keys"),
        py::arg("n_codewords"), py::arg("seed"), py::arg("prf_method"));
  m.def("grid_expand", &grid_expand, py::arg("key"), py::arg("prf_method"));
  m.attr("KEY_INTS") = py::int_(kKeyInts);
  m.attr("ENTRY_WORDS") = py::int_(kEntryWords);
  m.attr("PRF_DUMMY") = py::int_((int)PRF_DUMMY);
  m.attr("PRF_SALSA20") = py::int_((int)PRF_SALSA20);
  m.attr("PRF_CHACHA20") = py::int_((int)PRF_CHACHA20);
  m.attr("PRF_AES128") = py::int_((int)PRF_AES128);
}
