"""Multi-process sharded evaluation over gloo (CPU stand-in for the RCCL
path; same collective calls, same sharding math as the GPU path)."""

import os

import numpy as np
import torch
import torch.multiprocessing as mp
import pytest


def _worker(rank, world, port, q):
    import torch.distributed as td

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    td.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from gpudpf import DPF, ShardedDPF, _core

        torch.manual_seed(1234)
        N = 8192
        table = torch.randint(-(2**31), 2**31 - 1, (N, 16), dtype=torch.int64).to(
            torch.int32
        )
        # keys must be identical on every rank (they come from one client):
        # generate with a fixed seed
        ks = []
        idxs = [7, 4242, N - 1]
        for i in idxs:
            k1, k2 = _core.gen(i, N, b"dist-seed-%d" % i, DPF.PRF_SALSA20)
            ks.append((torch.from_numpy(k1), torch.from_numpy(k2)))

        sd = ShardedDPF(prf=DPF.PRF_SALSA20, device="cpu")
        sd.eval_init(table)
        a = sd.eval_cpu([k[0] for k in ks])
        b = sd.eval_cpu([k[1] for k in ks])
        rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32).numpy()
        gt = table[idxs, :].numpy()
        ok = bool(np.array_equal(rec, gt))
        q.put((rank, ok, ""))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        td.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4, 8])
def test_sharded_eval_matches_single(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29781 + world
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, q)) for r in range(world)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"


def test_shard_subkey_batch_matches_single():
    # one C++ call for the whole batch == per-key restriction
    from gpudpf import DPF, _core
    import numpy as np

    n, world = 1 << 14, 4
    prf = DPF.PRF_CHACHA20
    keys = []
    for i in range(8):
        k1, _ = _core.gen((i * 977) % n, n, b"sb-%d" % i, prf)
        keys.append(k1)
    kt = np.stack(keys)
    for rank in range(world):
        got = _core.shard_subkey_batch(kt, prf, rank, world)
        for i in range(8):
            want = _core.shard_subkey(keys[i], prf, rank, world)
            assert np.array_equal(got[i], want), (rank, i)
