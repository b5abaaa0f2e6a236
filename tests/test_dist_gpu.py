"""ShardedDPF.eval_gpu end-to-end on a real GPU: 2 ranks (gloo rendezvous,
both mapped to cuda:0 on a 1-GPU box) evaluate their shard on the GPU and
combine with the collective.  On an 8-GPU node the same code runs one rank
per GPU over RCCL."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, q):
    import torch.distributed as td

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    td.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from gpudpf import DPF, ShardedDPF, _core

        N = 1 << 15
        prf = DPF.PRF_AES128
        torch.manual_seed(7)
        table = torch.randint(-(2**31), 2**31 - 1, (N, 16), dtype=torch.int64).to(
            torch.int32
        )
        idxs = [5, 30000, 12345]
        ks = []
        for i in idxs:
            k1, k2 = _core.gen(i, N, b"dist-gpu-%d" % i, prf)
            ks.append((torch.from_numpy(k1), torch.from_numpy(k2)))

        sd = ShardedDPF(prf=prf, device="cuda:0")
        sd.eval_init(table)
        a = sd.eval_gpu([k[0] for k in ks])
        b = sd.eval_gpu([k[1] for k in ks])
        rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32).numpy()
        ok = bool(np.array_equal(rec, table[idxs, :].numpy()))
        q.put((rank, ok, ""))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        td.destroy_process_group()


def test_sharded_gpu_eval_two_ranks():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29791, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, f"rank {rank}: {err}"
