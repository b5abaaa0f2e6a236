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


def _digit_planes(x_u32_gpu):
    """[..., count] int32 tensor on GPU -> [4, count] int8 signed base-256
    digit planes (flattened over the input shape)."""
    count = x_u32_gpu.numel()
    out = torch.empty((4, count), dtype=torch.int8, device=x_u32_gpu.device)
    stream = torch.cuda.current_stream(x_u32_gpu.device).cuda_stream
    _hip.digits(x_u32_gpu.data_ptr(), out.data_ptr(), count, stream)
    return out


def pir_matmul_u32_stream(shares, table, out=None):
    """C = shares @ table mod 2^32, streaming the u32 table in place.

    shares: [M, K] int32, table: [K, N] int32 (both already on the GPU for
    the huge-table path).  Unlike pir_matmul_u32 this allocates NO copy of
    the table (no transpose, no digit planes) — the extra memory is the
    [M, N] output plus a <=1 GiB K-split partials scratch — so it serves
    tables sized to HBM capacity (200+ GB).  Batches > 16 are evaluated
    in 16-row chunks, each chunk re-streaming the table once: 16 is the
    bandwidth-bound register limit (4 uint4 accumulator sets per lane);
    re-streaming at the HBM line beats one instruction-bound pass, and
    genuinely compute-bound shapes belong to pir_matmul_u32 (MFMA)."""
    assert shares.dtype == torch.int32 and table.dtype == torch.int32
    M, K = shares.shape
    K2, N = table.shape
    assert K == K2
    dev = torch.device("cuda:0") if shares.device.type == "cpu" else shares.device
    a = shares.to(dev).contiguous()
    b = table.to(dev).contiguous()
    c = out if out is not None else torch.empty(
        (M, N), dtype=torch.int32, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    for lo in range(0, M, 16):
        hi = min(M, lo + 16)
        chunk = c[lo:hi]
        chunk.zero_()
        _hip.gemm_u32_stream(a[lo:hi].data_ptr(), b.data_ptr(),
                             chunk.data_ptr(), hi - lo, K, N, stream)
    return c


def pir_matmul_u32(shares, table):
    """C = shares @ table mod 2^32 on the MFMA matrix cores.

    shares: [M, K] int32 (e.g. one-hot DPF shares), table: [K, N] int32.
    Each u32 operand is decomposed into 4 signed base-256 digit planes and
    the product assembled from 10 i8-MFMA accumulator chains (see
    csrc/hip/gemm_u32.hip).  Exact mod 2^32.
    """
    assert shares.dtype == torch.int32 and table.dtype == torch.int32
    M, K = shares.shape
    K2, N = table.shape
    assert K == K2
    dev = torch.device("cuda:0") if shares.device.type == "cpu" else shares.device
    Mp, Np, Kp = -(-M // 64) * 64, -(-N // 16) * 16, -(-K // 64) * 64
    a = torch.zeros((Mp, Kp), dtype=torch.int32, device=dev)
    a[:M, :K] = shares.to(dev)
    bt = torch.zeros((Np, Kp), dtype=torch.int32, device=dev)
    bt[:N, :K] = table.to(dev).t()
    da = _digit_planes(a)
    dbt = _digit_planes(bt)
    c = torch.zeros((Mp, Np), dtype=torch.int32, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    _hip.gemm_u32_mfma(da.data_ptr(), dbt.data_ptr(), c.data_ptr(), Mp, Np,
                       Kp, stream)
    return c[:M, :N]
