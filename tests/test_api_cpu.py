"""DPF python API tests, CPU paths (mirrors the reference's python
integration tests dpf.py:139-204 as pytest)."""

import random

import numpy as np
import pytest
import torch

from gpudpf import DPF, _core


def test_cpu_dpf_one_hot():
    N = 1024
    dpf = DPF()
    K = 42
    k1, k2 = dpf.gen(K, N)
    v1 = dpf.eval_cpu([k1], one_hot_only=True)
    v2 = dpf.eval_cpu([k2], one_hot_only=True)
    rec = (v1 - v2).numpy()
    gt = np.zeros(rec.shape)
    gt[:, K] = 1
    assert np.linalg.norm(rec - gt) <= 1e-8


def test_cpu_dpf_with_table():
    N = 1024
    dpf = DPF(device="cpu")
    k1s, k2s, gt_indices = [], [], []
    for _ in range(16):
        indx = random.randint(0, N - 1)
        gt_indices.append(indx)
        k1, k2 = dpf.gen(indx, N)
        k1s.append(k1)
        k2s.append(k2)
    table = torch.arange(N * 16, dtype=torch.int32).reshape(N, 16)
    dpf.eval_init(table)
    a = dpf.eval_cpu(k1s)
    b = dpf.eval_cpu(k2s)
    rec = (a - b).numpy()
    gt = table[gt_indices, :].numpy()
    assert np.linalg.norm(rec - gt) <= 1e-8


def test_gen_validation():
    dpf = DPF()
    with pytest.raises(Exception):
        dpf.gen(1024, 1024)  # k >= n
    # non-power-of-two n is supported (padded domain); key domain is the
    # next power of two >= 128
    k1, _ = dpf.gen(0, 1000)
    assert k1[520] == 1024
    k1, _ = dpf.gen(0, 50)
    assert k1[520] == 128


def test_eval_init_accepts_extended_shapes():
    dpf = DPF(device="cpu")
    dpf.eval_init(torch.zeros((64, 16), dtype=torch.int32))   # < 128 rows: padded
    assert dpf._n_domain == 128
    dpf.eval_init(torch.zeros((100, 16), dtype=torch.int32))  # not pow2: padded
    assert dpf._n_domain == 128
    dpf.eval_init(torch.zeros((128, 17), dtype=torch.int32))  # wide entry
    assert dpf._entry_padded == 32


def test_key_size_constant():
    dpf = DPF()
    for n in [128, 8192, 1 << 20]:
        k1, _ = dpf.gen(1, n)
        assert int(np.prod(k1.shape)) * 4 == 2096


def test_prf_choice():
    for prf in [DPF.PRF_DUMMY, DPF.PRF_SALSA20, DPF.PRF_CHACHA20, DPF.PRF_AES128]:
        dpf = DPF(prf=prf)
        k1, k2 = dpf.gen(3, 128)
        v = (dpf.eval_cpu([k1], one_hot_only=True) -
             dpf.eval_cpu([k2], one_hot_only=True))
        assert v[0, 3] == 1 and int((v != 0).sum()) == 1


def test_repr():
    dpf = DPF(device="cpu")
    assert "uninitialized" in repr(dpf)
    dpf.eval_init(torch.zeros((128, 4), dtype=torch.int32))
    assert "entries=128" in repr(dpf) and "entry_size=4" in repr(dpf)


def test_non_pow2_table_cpu():
    N = 1000  # padded to 1024 domain
    dpf = DPF(prf=DPF.PRF_SALSA20, device="cpu")
    table = torch.arange(N * 4, dtype=torch.int32).reshape(N, 4)
    dpf.eval_init(table)
    idxs = [0, 999, 500]
    k1s, k2s = [], []
    for i in idxs:
        k1, k2 = dpf.gen(i, N)
        k1s.append(k1)
        k2s.append(k2)
    rec = (dpf.eval_cpu(k1s).to(torch.int64) -
           dpf.eval_cpu(k2s).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :])


def test_small_table_padded_to_128():
    dpf = DPF(device="cpu")
    table = torch.ones((50, 2), dtype=torch.int32)
    dpf.eval_init(table)
    k1, k2 = dpf.gen(10, 50)
    rec = (dpf.eval_cpu([k1]).to(torch.int64) -
           dpf.eval_cpu([k2]).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec[0], table[10])


def test_gen_batch():
    N = 2048
    dpf = DPF(prf=DPF.PRF_SALSA20, device="cpu")
    idxs = [0, 5, 2047, 1024]
    k1s, k2s = dpf.gen_batch(idxs, N)
    assert k1s.shape == (4, 524)
    v1 = dpf.eval_cpu(k1s, one_hot_only=True)
    v2 = dpf.eval_cpu(k2s, one_hot_only=True)
    rec = (v1 - v2).numpy()
    for i, a in enumerate(idxs):
        assert rec[i, a] == 1 and np.count_nonzero(rec[i]) == 1


def test_corrupt_key_rejected():
    dpf = DPF(device="cpu")
    garbage = torch.zeros(524, dtype=torch.int32)  # depth 0: invalid
    with pytest.raises(Exception):
        dpf.eval_cpu([garbage], one_hot_only=True)
    garbage[0] = 99  # depth out of range
    with pytest.raises(Exception):
        dpf.eval_cpu([garbage], one_hot_only=True)


def test_mixed_domain_batch_rejected():
    # A batch mixing key domains must raise, not silently evaluate at the
    # first key's depth (round-1 advisor finding).
    d = DPF(prf=DPF.PRF_SALSA20, device="cpu")
    ka, _ = d.gen(3, 1024)
    kb, _ = d.gen(3, 2048)
    with pytest.raises(Exception, match="domain"):
        d.eval_cpu([ka, kb], one_hot_only=True)


def test_large_domain_header_decode():
    # n is a u64 in the wire header: a 2^32-entry domain (depth 32) must
    # decode correctly, not truncate to the low 32 bits.
    k1, _ = _core.gen(123, 1 << 32, b"big", _core.PRF_DUMMY)
    d = DPF(prf=DPF.PRF_DUMMY, device="cpu")
    kt, n, depth = d._keys_tensor(torch.from_numpy(k1).unsqueeze(0))
    assert n == 1 << 32 and depth == 32


def test_compact_key_round_trip():
    d = DPF(prf=DPF.PRF_CHACHA20, device="cpu")
    for n, depth in ((128, 7), (1 << 14, 14)):
        k1, k2 = d.gen(n // 2, n)
        c = DPF.key_compact(k1)
        assert c.numel() * 4 == (3 + 4 * depth) * 16
        back = DPF.key_expand(c)
        assert torch.equal(back, k1)
        # expanded key evaluates identically
        a = d.eval_cpu([back], one_hot_only=True)
        b = d.eval_cpu([k1], one_hot_only=True)
        assert torch.equal(a, b)
    with pytest.raises(Exception):
        DPF.key_expand(torch.zeros(16, dtype=torch.int32))
