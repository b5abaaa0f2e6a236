"""GEMM128 tests: CPU reference self-consistency (vs python ints) and the
GPU kernel vs the CPU reference (the reference's 128-bit unit-test tier:
dpf_gpu/tests/test_128_bit.cu + matmul check)."""

import numpy as np
import pytest
import torch

from gpudpf import ops


def _to_u128(arr):
    # [.,.,4] int32 LE limbs -> python int matrix
    a = arr.astype(np.uint32).astype(object)
    return a[..., 0] + (a[..., 1] << 32) + (a[..., 2] << 64) + (a[..., 3] << 96)


def test_gemm128_cpu_reference_exact():
    rng = np.random.default_rng(7)
    M, N, K = 3, 2, 5
    a = rng.integers(-(2**31), 2**31 - 1, (M, K, 4), dtype=np.int64).astype(np.int32)
    bt = rng.integers(-(2**31), 2**31 - 1, (N, K, 4), dtype=np.int64).astype(np.int32)
    got = ops.gemm128_cpu(torch.from_numpy(a), torch.from_numpy(bt)).numpy()
    av = _to_u128(a)
    bv = _to_u128(bt)
    mask = (1 << 128) - 1
    for m in range(M):
        for n in range(N):
            want = 0
            for k in range(K):
                want = (want + int(av[m, k]) * int(bv[n, k])) & mask
            gotv = int(_to_u128(got)[m, n])
            assert gotv == want


@pytest.mark.gpu
def test_gemm128_gpu_matches_cpu():
    torch.manual_seed(3)
    for (M, N, K) in [(32, 16, 4096), (7, 5, 1000), (64, 16, 131072)]:
        a = torch.randint(-(2**31), 2**31 - 1, (M, K, 4), dtype=torch.int64).to(
            torch.int32
        )
        bt = torch.randint(-(2**31), 2**31 - 1, (N, K, 4), dtype=torch.int64).to(
            torch.int32
        )
        got = ops.gemm128(a, bt).cpu()
        want = ops.gemm128_cpu(a, bt)
        assert torch.equal(got, want), (M, N, K)
