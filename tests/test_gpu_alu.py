"""Dedicated GPU 128-bit ALU and PRF-core unit tests (the analog of the
reference's dpf_gpu/tests/test_128_bit.cu:192-200 single-thread probes):
the device add128 / mul128 / each PRF pair/single/low variant is compared
elementwise against the CPU core over random vectors, so a PRF-core bug is
caught here rather than only by end-to-end reconstruction."""

import numpy as np
import pytest
import torch

from gpudpf import DPF, _core

try:
    from gpudpf import _hip
except ImportError:  # pragma: no cover
    _hip = None

pytestmark = pytest.mark.gpu

COUNT = 4096
MASK128 = (1 << 128) - 1


def _rand_u128_words(rng, count):
    """[count, 4] uint32 limb array (little-endian words)."""
    return rng.integers(0, 1 << 32, size=(count, 4), dtype=np.uint64).astype(
        np.uint32
    )


def _to_int(words):
    """[4] limbs -> python int."""
    return int(words[0]) | (int(words[1]) << 32) | (int(words[2]) << 64) | (
        int(words[3]) << 96
    )


def _to_words(v):
    return np.array([(v >> (32 * i)) & 0xFFFFFFFF for i in range(4)],
                    dtype=np.uint32)


def _dev(arr):
    return torch.from_numpy(arr.view(np.int32)).to("cuda:0").contiguous()


def test_device_add128_mul128():
    rng = np.random.default_rng(42)
    a = _rand_u128_words(rng, COUNT)
    b = _rand_u128_words(rng, COUNT)
    # force carry-chain edge cases into the vector
    a[0] = _to_words(MASK128)
    b[0] = _to_words(1)
    a[1] = _to_words((1 << 64) - 1)
    b[1] = _to_words(1)
    a[2] = _to_words(MASK128)
    b[2] = _to_words(MASK128)
    ga, gb = _dev(a), _dev(b)
    add_out = torch.empty((COUNT, 4), dtype=torch.int32, device="cuda:0")
    mul_out = torch.empty_like(add_out)
    stream = torch.cuda.current_stream().cuda_stream
    _hip.probe_alu(ga.data_ptr(), gb.data_ptr(), add_out.data_ptr(),
                   mul_out.data_ptr(), COUNT, stream)
    add_np = add_out.cpu().numpy().view(np.uint32)
    mul_np = mul_out.cpu().numpy().view(np.uint32)
    for i in range(COUNT):
        x, y = _to_int(a[i]), _to_int(b[i])
        assert _to_int(add_np[i]) == (x + y) & MASK128, f"add128 row {i}"
        assert _to_int(mul_np[i]) == (x * y) & MASK128, f"mul128 row {i}"


@pytest.mark.parametrize(
    "prf",
    [DPF.PRF_DUMMY, DPF.PRF_SALSA20, DPF.PRF_CHACHA20, DPF.PRF_AES128],
)
def test_device_prf_matches_cpu_core(prf):
    rng = np.random.default_rng(1234 + prf)
    seeds = _rand_u128_words(rng, COUNT)
    seeds[0] = _to_words(0)
    seeds[1] = _to_words(MASK128)
    gs = _dev(seeds)
    dev = "cuda:0"
    bufs = [torch.empty((COUNT, 4), dtype=torch.int32, device=dev)
            for _ in range(4)]
    lows = [torch.empty(COUNT, dtype=torch.int32, device=dev)
            for _ in range(2)]
    aes_ptr = _hip.ensure_aes_tables(0) if prf == DPF.PRF_AES128 else 0
    stream = torch.cuda.current_stream().cuda_stream
    _hip.probe_prf(gs.data_ptr(), aes_ptr, bufs[0].data_ptr(),
                   bufs[1].data_ptr(), bufs[2].data_ptr(), bufs[3].data_ptr(),
                   lows[0].data_ptr(), lows[1].data_ptr(), COUNT, prf, stream)
    pair0, pair1, single0, single1 = (b.cpu().numpy().view(np.uint32)
                                      for b in bufs)
    low0, low1 = (l.cpu().numpy().view(np.uint32) for l in lows)
    for i in range(COUNT):
        s_lo = int(seeds[i][0]) | (int(seeds[i][1]) << 32)
        s_hi = int(seeds[i][2]) | (int(seeds[i][3]) << 32)
        for pos, pair, single, low in ((0, pair0, single0, low0),
                                       (1, pair1, single1, low1)):
            lo, hi = _core.prf(prf, s_lo, s_hi, pos)
            want = (int(hi) << 64) | int(lo)
            got_pair = _to_int(pair[i])
            got_single = _to_int(single[i])
            assert got_pair == want, (
                f"prf={prf} pos={pos} row {i}: pair-core {got_pair:#x} != "
                f"cpu {want:#x}")
            assert got_single == want, (
                f"prf={prf} pos={pos} row {i}: single-core mismatch")
            assert int(low[i]) == want & 0xFFFFFFFF, (
                f"prf={prf} pos={pos} row {i}: low-variant mismatch")
