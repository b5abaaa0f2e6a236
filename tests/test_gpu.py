"""GPU correctness tests: HIP fused/expand/naive kernels vs the CPU core
oracle (reference strategy: CPU keygen <-> GPU eval cross-validation,
dpf.py:206-243, plus DUMMY full-output oracles, utils.h:152-209)."""

import random

import numpy as np
import pytest
import torch

from gpudpf import DPF, _core

pytestmark = pytest.mark.gpu

PRFS = [DPF.PRF_DUMMY, DPF.PRF_SALSA20, DPF.PRF_CHACHA20, DPF.PRF_AES128]


def _roundtrip(n, batch, entrysize, prf, seed=0):
    random.seed(seed)
    dpf = DPF(prf=prf)
    k1s, k2s, gt_indices = [], [], []
    for _ in range(batch):
        indx = random.randint(0, n - 1)
        gt_indices.append(indx)
        k1, k2 = dpf.gen(indx, n)
        k1s.append(k1)
        k2s.append(k2)
    table = torch.randint(-(2**31), 2**31 - 1, (n, entrysize), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    a = dpf.eval_gpu(k1s)
    b = dpf.eval_gpu(k2s)
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32).numpy()
    gt = table[gt_indices, :].numpy()
    assert np.array_equal(rec, gt), (n, batch, entrysize, prf)


@pytest.mark.parametrize("prf", PRFS)
def test_gpu_fused_all_prfs(prf):
    _roundtrip(8192, 64, 16, prf)


def test_gpu_fused_sweep():
    for n in [128, 256, 512, 1024, 8192, 65536]:
        _roundtrip(n, random.randint(1, 300), random.randint(1, 16), DPF.PRF_SALSA20,
                   seed=n)


def test_gpu_fused_large_batch_chunking(monkeypatch):
    # force multiple kernel launches per eval_gpu call
    monkeypatch.setattr(DPF, "MAX_LAUNCH_BATCH", 256)
    _roundtrip(1024, 600, 5, DPF.PRF_CHACHA20)


def test_gpu_onehot_matches_cpu():
    n = 8192
    for prf in PRFS:
        dpf = DPF(prf=prf)
        k1, k2 = dpf.gen(1234, n)
        table = torch.zeros((n, 1), dtype=torch.int32)
        dpf.eval_init(table)
        got = dpf.eval_gpu([k1, k2], one_hot_only=True).numpy()
        want1 = _core.expand(k1.numpy(), prf)
        want2 = _core.expand(k2.numpy(), prf)
        assert np.array_equal(got[0], want1), prf
        assert np.array_equal(got[1], want2), prf


def test_gpu_bfs_strategy_matches_cpu():
    # level-synchronized breadth-first expansion (natural-order output)
    # must agree with the CPU core for every PRF, including a deep-enough
    # tree to cross several frontier ping-pongs
    for n in (512, 1 << 14):
        for prf in PRFS:
            dpf = DPF(prf=prf)
            k1, k2 = dpf.gen(n // 3, n)
            table = torch.zeros((n, 1), dtype=torch.int32)
            dpf.eval_init(table)
            got = dpf.eval_gpu([k1, k2], one_hot_only=True,
                               strategy="bfs").numpy()
            want1 = _core.expand(k1.numpy(), prf)
            want2 = _core.expand(k2.numpy(), prf)
            assert np.array_equal(got[0], want1), (n, prf)
            assert np.array_equal(got[1], want2), (n, prf)


def test_gpu_pipelined_server_interleaved_batches():
    # double-buffered serving: interleaved submits of DIFFERENT batches
    # must return each batch's own correct shares
    from gpudpf.serving import PipelinedServer

    n, batch = 16384, 64
    dpf = DPF(prf=DPF.PRF_SALSA20)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16),
                          dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)
    srv = PipelinedServer(dpf, batch, depth=2)
    batches = []
    for r in range(6):
        ks = torch.stack([dpf.gen((r * 1000 + i * 13) % n, n)[0]
                          for i in range(batch)])
        batches.append(ks)
    pending = []
    results = []
    for ks in batches:
        pending.append((srv.submit(ks), ks))
        if len(pending) >= 2:
            h, k = pending.pop(0)
            results.append((srv.collect(h), k))
    while pending:
        h, k = pending.pop(0)
        results.append((srv.collect(h), k))
    for got, ks in results:
        want = dpf.eval_gpu(ks)
        assert torch.equal(got, want)


def test_gpu_two_stage_server_pipelined():
    # wide-entry pipelined serving: expansion of batch i+1 overlaps the
    # GEMM of batch i; results must match the plain two-stage path
    from gpudpf.serving import TwoStageServer

    n, e, batch = 8192, 48, 8
    dpf = DPF(prf=DPF.PRF_CHACHA20)
    table = torch.randint(-(2**31), 2**31 - 1, (n, e),
                          dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)
    srv = TwoStageServer(dpf, batch)
    batches = [torch.stack([dpf.gen((r * 311 + i * 7) % n, n)[0]
                            for i in range(batch)]) for r in range(5)]
    pending, results = [], []
    for ks in batches:
        pending.append((srv.submit(ks), ks))
        if len(pending) >= 2:
            h, k = pending.pop(0)
            results.append((srv.collect(h), k))
    while pending:
        h, k = pending.pop(0)
        results.append((srv.collect(h), k))
    for got, ks in results:
        assert torch.equal(got, dpf.eval_gpu(ks))


def test_gpu_coop_strategy_matches_fused():
    # grid-wide cooperative kernel (one cooperative launch per key, grid
    # sync per level): fused output must equal the production path, and
    # the one-hot output must match the CPU core in natural order
    n = 1 << 14
    for prf in PRFS:
        dpf = DPF(prf=prf)
        table = torch.randint(-(2**31), 2**31 - 1, (n, 16),
                              dtype=torch.int64).to(torch.int32)
        dpf.eval_init(table)
        k1, k2 = dpf.gen(9999, n)
        kt = torch.stack([k1, k2])
        want = dpf.eval_gpu(kt)
        got = dpf.eval_gpu(kt, strategy="coop")
        assert torch.equal(got, want), prf
        oh = dpf.eval_gpu([k1], one_hot_only=True, strategy="coop").numpy()
        assert np.array_equal(oh[0], _core.expand(k1.numpy(), prf)), prf


def test_gpu_naive_kernel_oracle():
    from gpudpf import _hip

    n, batch = 8192, 8
    for prf in [DPF.PRF_DUMMY, DPF.PRF_AES128]:
        dpf = DPF(prf=prf)
        keys = []
        for i in range(batch):
            k1, _ = dpf.gen((i * 977) % n, n)
            keys.append(k1)
        kt = torch.stack(keys).contiguous().to("cuda:0")
        out = torch.empty((batch, n), dtype=torch.int32, device="cuda:0")
        aes_ptr = _hip.ensure_aes_tables(0)
        stream = torch.cuda.current_stream().cuda_stream
        depth = n.bit_length() - 1
        _hip.eval_naive(kt.data_ptr(), out.data_ptr(), aes_ptr, batch, n, depth,
                        prf, stream)
        got = out.cpu().numpy()
        for i in range(batch):
            want = _core.expand(keys[i].numpy(), prf)
            assert np.array_equal(got[i], want), (prf, i)


def test_gpu_fused_matches_cpu_fused_oracle():
    # direct single-server value check (not just the a-b reconstruction):
    # catches compensating errors identical across the two servers
    n = 16384
    prf = DPF.PRF_SALSA20
    dpf = DPF(prf=prf)
    k1, _ = dpf.gen(777, n)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    got = dpf.eval_gpu([k1]).numpy()[0]
    want = _core.eval_fused_cpu(k1.numpy(), table.numpy(), prf)
    assert np.array_equal(got.astype(np.uint32), want.astype(np.uint32))


def test_gpu_large_domain_salsa():
    n = 1 << 18
    dpf = DPF(prf=DPF.PRF_SALSA20)
    alpha = 123456
    k1, k2 = dpf.gen(alpha, n)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    a = dpf.eval_gpu([k1])
    b = dpf.eval_gpu([k2])
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32).numpy()
    assert np.array_equal(rec[0], table[alpha].numpy())


def test_gpu_sharded_single_process():
    # simulate 4-way sharding in one process (no collective): partial sums
    # over sub-keys must add up to the unsharded result
    n = 1 << 16
    world = 4
    prf = DPF.PRF_CHACHA20
    dpf = DPF(prf=prf)
    alpha = 54321
    k1, k2 = dpf.gen(alpha, n)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    full_a = dpf.eval_gpu([k1])

    acc = torch.zeros((1, 16), dtype=torch.int32)
    for r in range(world):
        sub = torch.from_numpy(_core.shard_subkey(k1.numpy(), prf, r, world))
        d = DPF(prf=prf)
        d.eval_init(table[r::world].contiguous())
        acc = (acc.to(torch.int64) + d.eval_gpu([sub]).to(torch.int64)).to(torch.int32)
    assert np.array_equal(acc.numpy(), full_a.numpy())


def test_gpu_perf_smoke():
    import time

    n, batch = 65536, 512
    dpf = DPF(prf=DPF.PRF_AES128)
    k1, _ = dpf.gen(1, n)
    keys = torch.stack([k1] * batch)
    table = torch.zeros((n, 16), dtype=torch.int32)
    dpf.eval_init(table)
    dpf.eval_gpu(keys)  # warmup
    torch.cuda.synchronize()
    t0 = time.time()
    reps = 5
    for _ in range(reps):
        dpf.eval_gpu(keys)
    torch.cuda.synchronize()
    dt = time.time() - t0
    print("AES128 n=%d: %.0f dpfs/sec" % (n, batch * reps / dt))


def test_gpu_two_stage_strategy_matches_fused():
    n = 16384
    for prf in [DPF.PRF_SALSA20, DPF.PRF_AES128]:
        dpf = DPF(prf=prf)
        alpha = 777
        k1, k2 = dpf.gen(alpha, n)
        table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
            torch.int32
        )
        dpf.eval_init(table)
        fused = dpf.eval_gpu([k1, k2])
        two = dpf.eval_gpu([k1, k2], strategy="two_stage")
        assert torch.equal(fused, two), prf
        rec = (two[0].to(torch.int64) - two[1].to(torch.int64)).to(torch.int32)
        assert torch.equal(rec, table[alpha])


def test_gpu_wide_entries_and_non_pow2():
    # entries wider than 16 words route through the MFMA two-stage path;
    # non-power-of-two n pads to the next domain
    n, e = 5000, 40
    dpf = DPF(prf=DPF.PRF_CHACHA20)
    table = torch.randint(-(2**31), 2**31 - 1, (n, e), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    idxs = [0, 4999, 1234]
    k1s, k2s = [], []
    for i in idxs:
        k1, k2 = dpf.gen(i, n)
        k1s.append(k1)
        k2s.append(k2)
    a = dpf.eval_gpu(k1s)
    b = dpf.eval_gpu(k2s)
    assert a.shape == (3, e)
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :])


def test_gpu_fused_deterministic():
    # the j-split combines segment partials with atomics; mod-2^32 addition
    # is commutative/associative, so results must be bitwise reproducible
    n = 1 << 17
    dpf = DPF(prf=DPF.PRF_SALSA20)
    k1, _ = dpf.gen(4242, n)
    keys = torch.stack([k1] * 64)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    a = dpf.eval_gpu(keys)
    for _ in range(3):
        assert torch.equal(dpf.eval_gpu(keys), a)


def test_gpu_graphed_server_matches_eval():
    from gpudpf.serving import GraphedServer

    n, batch = 16384, 64
    dpf = DPF(prf=DPF.PRF_AES128)
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    srv = GraphedServer(dpf, batch)
    keys = []
    for i in range(batch):
        k1, _ = dpf.gen((i * 37) % n, n)
        keys.append(k1)
    kt = torch.stack(keys)
    want = dpf.eval_gpu(kt)
    got = srv.eval(kt)
    assert torch.equal(got, want)
    # replay with different keys reuses the same graph
    keys2 = [dpf.gen((i * 11) % n, n)[0] for i in range(batch)]
    kt2 = torch.stack(keys2)
    assert torch.equal(srv.eval(kt2), dpf.eval_gpu(kt2))


def test_gpu_graphed_server_wide_entries():
    # wide path: captured expand + streaming-GEMM graph, incl. >64-batch
    # (two GEMM launches in one graph)
    from gpudpf.serving import GraphedServer

    n, e, batch = 8192, 40, 96
    dpf = DPF(prf=DPF.PRF_CHACHA20)
    table = torch.randint(-(2**31), 2**31 - 1, (n, e), dtype=torch.int64).to(
        torch.int32)
    dpf.eval_init(table)
    srv = GraphedServer(dpf, batch)
    assert srv.wide
    keys = torch.stack([dpf.gen((i * 53) % n, n)[0] for i in range(batch)])
    want = dpf.eval_gpu(keys)
    got = srv.eval(keys)
    assert torch.equal(got, want)
    keys2 = torch.stack([dpf.gen((i * 7 + 1) % n, n)[0] for i in range(batch)])
    assert torch.equal(srv.eval(keys2), dpf.eval_gpu(keys2))


def test_gpu_graphed_server_scratch_regrow_safe():
    # Capture a graph at a small domain, then run a much deeper eval that
    # forces the shared scratch buffer to grow: the captured graph must
    # still replay correctly (round-1 advisor finding — the old code freed
    # the captured allocation).
    from gpudpf.serving import GraphedServer

    n_small, batch = 16384, 32
    dpf = DPF(prf=DPF.PRF_SALSA20)
    table = torch.randint(-(2**31), 2**31 - 1, (n_small, 16),
                          dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)
    srv = GraphedServer(dpf, batch)
    keys = torch.stack([dpf.gen(i % n_small, n_small)[0]
                        for i in range(batch)])
    want = srv.eval(keys)

    # deep eval on a second instance grows the per-device scratch
    n_big = 1 << 22
    dpf2 = DPF(prf=DPF.PRF_SALSA20)
    t2 = torch.zeros((n_big, 16), dtype=torch.int32)
    dpf2.eval_init(t2)
    k1, _ = dpf2.gen(5, n_big)
    dpf2.eval_gpu([k1])

    # the original graph still replays against live memory
    got = srv.eval(keys)
    assert torch.equal(got, want)


def test_gpu_deep_tree_correctness():
    # n=2^22: DS=14 -> 7 global-scratch stack levels + 4 LDS + 2 register
    # levels all exercised; reconstruction vs ground truth
    n = 1 << 22
    dpf = DPF(prf=DPF.PRF_AES128)
    alphas = [0, n - 1, 123456, 3999999]
    table = torch.randint(-(2**31), 2**31 - 1, (n, 16), dtype=torch.int64).to(
        torch.int32
    )
    dpf.eval_init(table)
    k1s, k2s = [], []
    for a in alphas:
        k1, k2 = dpf.gen(a, n)
        k1s.append(k1)
        k2s.append(k2)
    ra = dpf.eval_gpu(k1s)
    rb = dpf.eval_gpu(k2s)
    rec = (ra.to(torch.int64) - rb.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[alphas, :])


def test_gpu_onehot_deep_tree():
    # one-hot expansion kernel with global-scratch stack levels (n=2^20)
    n = 1 << 20
    dpf = DPF(prf=DPF.PRF_SALSA20)
    k1, _ = dpf.gen(987654, n)
    dpf.eval_init(torch.zeros((n, 1), dtype=torch.int32))
    got = dpf.eval_gpu([k1], one_hot_only=True).numpy()[0]
    want = _core.expand(k1.numpy(), DPF.PRF_SALSA20)
    assert np.array_equal(got, want)
