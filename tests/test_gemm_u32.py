"""MFMA mod-2^32 GEMM tests (GPU): exactness vs a wrapping int64 numpy
reference, including an accumulator-overflow case that proves i32 MFMA
accumulation wraps (not saturates)."""

import numpy as np
import pytest
import torch

from gpudpf import ops

pytestmark = pytest.mark.gpu


def _ref_mod32(a, b):
    # numpy int64 matmul wraps mod 2^64; truncating to uint32 is exact mod 2^32
    prod = a.astype(np.int64) @ b.astype(np.int64)
    return prod.astype(np.uint32).astype(np.int32)


def test_gemm_u32_exact_random():
    rng = np.random.default_rng(11)
    for (M, N, K) in [(64, 16, 64), (128, 16, 4096), (100, 10, 1000),
                      (512, 16, 65536)]:
        a = rng.integers(-(2**31), 2**31 - 1, (M, K), dtype=np.int64).astype(
            np.int32
        )
        b = rng.integers(-(2**31), 2**31 - 1, (K, N), dtype=np.int64).astype(
            np.int32
        )
        got = ops.pir_matmul_u32(torch.from_numpy(a), torch.from_numpy(b)).cpu()
        want = torch.from_numpy(_ref_mod32(a, b))
        assert torch.equal(got, want), (M, N, K)


def test_gemm_u32_accumulator_wraps():
    # worst-case digit magnitudes with K deep enough that the per-pair i32
    # MFMA accumulator exceeds 2^31: 128*128*2^18 = 2^32
    M, N, K = 64, 16, 1 << 18
    a = np.full((M, K), 0x80808080, dtype=np.uint32).astype(np.int32)
    b = np.full((K, N), 0x80808080, dtype=np.uint32).astype(np.int32)
    got = ops.pir_matmul_u32(torch.from_numpy(a), torch.from_numpy(b)).cpu()
    want = torch.from_numpy(_ref_mod32(a, b))
    assert torch.equal(got, want)


def test_gemm_u32_one_hot_pir_identity():
    # one-hot share rows select table rows exactly
    M, N, K = 64, 16, 8192
    rng = np.random.default_rng(5)
    b = rng.integers(-(2**31), 2**31 - 1, (K, N), dtype=np.int64).astype(np.int32)
    a = np.zeros((M, K), dtype=np.int32)
    picks = rng.integers(0, K, M)
    a[np.arange(M), picks] = 1
    got = ops.pir_matmul_u32(torch.from_numpy(a), torch.from_numpy(b)).cpu()
    assert np.array_equal(got.numpy(), b[picks])
