"""Standalone GPU ops (research-harness components)."""

import torch

from gpudpf import _core

try:
    from gpudpf import _hip
except ImportError:  # pragma: no cover
    _hip = None


def gemm128(a, bt):
    """Exact GEMM over Z_2^128: a [M,K,4] int32 (u128 little-endian limbs),
    bt [N,K,4] int32 -> [M,N,4] int32, computed on the GPU.

    This is the reference's standalone GEMM128 research kernel
    (dpf_gpu/matmul/matmul.cu) as a CDNA4 HIP kernel with K split across
    blockIdx.z and a mod-2^128 reduction kernel."""
    assert a.dtype == torch.int32 and bt.dtype == torch.int32
    assert a.dim() == 3 and bt.dim() == 3 and a.shape[2] == 4 and bt.shape[2] == 4
    assert a.shape[1] == bt.shape[1]
    M, K = a.shape[0], a.shape[1]
    N = bt.shape[0]
    dev = torch.device("cuda:0") if a.device.type == "cpu" else a.device
    a_g = a.to(dev).contiguous()
    b_g = bt.to(dev).contiguous()
    c_g = torch.empty((M, N, 4), dtype=torch.int32, device=dev)
    ksplit = _hip.gemm128_ksplit(M, N, K)
    scratch = torch.empty((ksplit, M, N, 4), dtype=torch.int32, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    _hip.gemm128(a_g.data_ptr(), b_g.data_ptr(), c_g.data_ptr(),
                 scratch.data_ptr(), M, N, K, stream)
    return c_g


def gemm128_cpu(a, bt):
    """CPU reference for gemm128 (exact, single-threaded)."""
    import numpy as np

    return torch.from_numpy(
        np.asarray(_core.gemm128_cpu(a.numpy(), bt.numpy()))
    )
