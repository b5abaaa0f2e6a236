"""CPU core correctness: keygen/eval reconstruction, serialization, layout
permutation, sharding.  Mirrors the reference's test strategy (SURVEY.md
§4) as pytest."""

import random

import numpy as np
import pytest

from gpudpf import _core

PRFS = [_core.PRF_DUMMY, _core.PRF_SALSA20, _core.PRF_CHACHA20, _core.PRF_AES128]


def gen(alpha, n, prf, seed=b"test-seed"):
    return _core.gen(alpha, n, seed, prf)


@pytest.mark.parametrize("prf", PRFS)
def test_one_hot_reconstruction_full_domain(prf):
    n = 512
    alpha = random.randrange(n)
    k1, k2 = gen(alpha, n, prf)
    a = _core.expand(k1, prf).astype(np.int64)
    b = _core.expand(k2, prf).astype(np.int64)
    rec = (a - b).astype(np.int32)
    expected = np.zeros(n, dtype=np.int32)
    expected[alpha] = 1
    assert np.array_equal(rec, expected)


def test_large_domain_dummy():
    n = 1 << 20
    alpha = 123457
    k1, k2 = gen(alpha, n, _core.PRF_DUMMY)
    a = _core.expand(k1, _core.PRF_DUMMY)
    b = _core.expand(k2, _core.PRF_DUMMY)
    rec = (a.astype(np.int64) - b.astype(np.int64)).astype(np.int32)
    assert rec[alpha] == 1
    assert np.count_nonzero(rec) == 1


@pytest.mark.parametrize("prf", PRFS)
def test_eval_point_matches_expand(prf):
    n = 256
    alpha = 77
    k1, _ = gen(alpha, n, prf)
    full = _core.expand(k1, prf)
    for idx in [0, 1, alpha, n - 1, 128]:
        low = _core.eval_point_low(k1, idx, prf)
        assert np.uint32(low) == np.uint32(full[idx])


def test_key_wire_format():
    n = 1024
    k1, k2 = gen(5, n, _core.PRF_AES128)
    assert k1.shape == (524,) and k1.dtype == np.int32
    # slot 0 = depth, slot 130 = n (u128 slots, low word first)
    assert k1[0] == 10 and k1[520] == n
    assert k2[0] == 10 and k2[520] == n
    # 2096-byte key independent of n (reference README.md:119)
    assert k1.nbytes == 2096


def test_deterministic_given_seed():
    a1 = _core.gen(42, 4096, b"abc", _core.PRF_SALSA20)
    a2 = _core.gen(42, 4096, b"abc", _core.PRF_SALSA20)
    assert np.array_equal(a1[0], a2[0]) and np.array_equal(a1[1], a2[1])
    b1 = _core.gen(42, 4096, b"abd", _core.PRF_SALSA20)
    assert not np.array_equal(a1[0], b1[0])


def test_wire_format_depth_limit():
    # The 524-int wire format holds 2*depth cw slots (64 max): depth <= 32
    # is a hard limit and must raise, never silently truncate.
    k1, k2 = _core.gen(7, 1 << 32, b"depth32", _core.PRF_DUMMY)  # max ok
    assert int(k1[0]) == 32
    with pytest.raises(Exception):
        _core.gen(7, 1 << 33, b"depth33", _core.PRF_DUMMY)


def test_keygen_csprng_stream():
    # Keygen randomness is AES-CTR (not mt19937): correction words of keys
    # from different seeds share no structure, and a one-bit seed change
    # flips ~half the cw bits (sanity, not a statistical proof).
    a = _core.gen(0, 4096, b"\x00" * 16, _core.PRF_DUMMY)[0]
    b = _core.gen(0, 4096, b"\x01" + b"\x00" * 15, _core.PRF_DUMMY)[0]
    cw_a = np.asarray(a[4:488], dtype=np.uint32)
    cw_b = np.asarray(b[4:488], dtype=np.uint32)
    used = cw_a != 0  # only populated slots
    diff = np.unpackbits(
        (cw_a[used] ^ cw_b[used]).view(np.uint8)
    ).mean()
    assert 0.4 < diff < 0.6


def test_fused_cpu_oracle():
    n, e = 2048, 16
    alpha = 999
    prf = _core.PRF_CHACHA20
    k1, k2 = gen(alpha, n, prf)
    table = np.random.randint(-(2**31), 2**31 - 1, size=(n, e), dtype=np.int64).astype(
        np.int32
    )
    a = _core.eval_fused_cpu(k1, table, prf).astype(np.int64)
    b = _core.eval_fused_cpu(k2, table, prf).astype(np.int64)
    rec = (a - b).astype(np.int32)
    assert np.array_equal(rec, table[alpha])


@pytest.mark.parametrize("n", [128, 512, 8192, 1 << 14])
def test_leaf_perm_bijection_and_slab_property(n):
    zlog = _core.zlog_for_depth(n.bit_length() - 1)
    perm = np.asarray(_core.leaf_perm_table(n, zlog))
    assert sorted(perm.tolist()) == list(range(n))
    # slab property: the DFS emits pairs (row, row+1) per thread; natural
    # indices mapping to rows 2m and 2m+1 must share all bits except the
    # last-consumed one (bit depth-1)
    depth = n.bit_length() - 1
    inv = np.empty(n, dtype=np.int64)
    inv[perm] = np.arange(n)
    for m in range(0, min(n, 512), 2):
        i0, i1 = inv[m], inv[m + 1]
        assert i1 - i0 == 1 << (depth - 1)


def test_expand_batch_threads():
    n = 1024
    prf = _core.PRF_SALSA20
    keys = []
    alphas = []
    for _ in range(8):
        a = random.randrange(n)
        alphas.append(a)
        k1, k2 = gen(a, n, prf, seed=bytes([a & 0xFF]) * 4)
        keys.append((k1, k2))
    out1 = np.asarray(_core.expand_batch([k for k, _ in keys], prf, 1))
    out4 = np.asarray(_core.expand_batch([k for k, _ in keys], prf, 4))
    assert np.array_equal(out1, out4)
    outb = np.asarray(_core.expand_batch([k for _, k in keys], prf, 4))
    rec = (out1.astype(np.int64) - outb.astype(np.int64)).astype(np.int32)
    for i, a in enumerate(alphas):
        assert rec[i, a] == 1
        assert np.count_nonzero(rec[i]) == 1


@pytest.mark.parametrize("world", [2, 4, 8])
def test_shard_subkey(world):
    n = 1 << 13
    prf = _core.PRF_SALSA20
    alpha = random.randrange(n)
    k1, k2 = gen(alpha, n, prf)
    full1 = _core.expand(k1, prf)
    full2 = _core.expand(k2, prf)
    for rank in range(world):
        s1 = _core.shard_subkey(k1, prf, rank, world)
        s2 = _core.shard_subkey(k2, prf, rank, world)
        assert s1[0] == (n.bit_length() - 1) - (world.bit_length() - 1)
        e1 = _core.expand(s1, prf)
        e2 = _core.expand(s2, prf)
        # rank owns natural rows idx % world == rank, local index idx//world
        assert np.array_equal(e1, full1[rank::world])
        assert np.array_equal(e2, full2[rank::world])


@pytest.mark.parametrize("prf", [_core.PRF_DUMMY, _core.PRF_AES128])
def test_grid_sqrt_n_construction(prf):
    n_keys, n_codewords = 64, 32
    n = n_keys * n_codewords
    alpha = 777
    k0, k1 = _core.grid_gen(alpha, n_keys, n_codewords, b"grid-seed", prf)
    a = _core.grid_expand(k0, prf).astype(np.int64)
    b = _core.grid_expand(k1, prf).astype(np.int64)
    rec = (a - b).astype(np.int32)
    assert rec[alpha] == 1
    assert np.count_nonzero(rec) == 1
    # communication size: n_keys seeds + 2*n_codewords codewords
    assert k0["seeds"].shape == (n_keys, 4)
    assert k0["cw_even"].shape == (n_codewords, 4)
