"""Streaming mod-2^32 GEMM tests (GPU): exactness of the in-place
u32-table kernel (csrc/hip/gemm_stream.hip) vs a wrapping int64 numpy
reference, across batch-template buckets, ragged column counts, and the
>64-batch python chunking path."""

import numpy as np
import pytest
import torch

from gpudpf import ops

pytestmark = pytest.mark.gpu


def _ref_mod32(a, b):
    prod = a.astype(np.int64) @ b.astype(np.int64)
    return prod.astype(np.uint32).astype(np.int32)


@pytest.mark.parametrize(
    "M,K,N",
    [
        (1, 4096, 224),    # B=4 bucket, non-pow2 wide entry
        (3, 1 << 15, 64),
        (8, 1 << 14, 256),
        (13, 1 << 13, 100),  # ragged N (tail columns masked)
        (32, 1 << 12, 16),   # narrow entries still exact
        (64, 1 << 13, 320),  # two column tiles
        (100, 1 << 12, 48),  # > 64: python chunking path
    ],
)
def test_gemm_stream_exact(M, K, N):
    rng = np.random.default_rng(M * 1000 + N)
    a = rng.integers(-(2**31), 2**31 - 1, (M, K), dtype=np.int64).astype(
        np.int32)
    b = rng.integers(-(2**31), 2**31 - 1, (K, N), dtype=np.int64).astype(
        np.int32)
    got = ops.pir_matmul_u32_stream(torch.from_numpy(a),
                                    torch.from_numpy(b)).cpu()
    want = torch.from_numpy(_ref_mod32(a, b))
    assert torch.equal(got, want), (M, K, N)


def test_gemm_stream_accumulator_wraps():
    M, K, N = 8, 1 << 18, 32
    a = np.full((M, K), 0x80808080, dtype=np.uint32).astype(np.int32)
    b = np.full((K, N), 0x80808080, dtype=np.uint32).astype(np.int32)
    got = ops.pir_matmul_u32_stream(torch.from_numpy(a),
                                    torch.from_numpy(b)).cpu()
    want = torch.from_numpy(_ref_mod32(a, b))
    assert torch.equal(got, want)


def test_gemm_stream_matches_mfma_path():
    # the two GEMM backends must agree bit-for-bit (dispatch equivalence)
    M, K, N = 128, 1 << 14, 32
    rng = np.random.default_rng(77)
    a = rng.integers(-(2**31), 2**31 - 1, (M, K), dtype=np.int64).astype(
        np.int32)
    b = rng.integers(-(2**31), 2**31 - 1, (K, N), dtype=np.int64).astype(
        np.int32)
    s = ops.pir_matmul_u32_stream(torch.from_numpy(a), torch.from_numpy(b))
    m = ops.pir_matmul_u32(torch.from_numpy(a), torch.from_numpy(b))
    assert torch.equal(s.cpu(), m.cpu())
