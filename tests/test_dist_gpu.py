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


def _nccl_worker(port, collective, q):
    """world=1 RCCL process group on the single GPU: executes the real
    nccl(=RCCL) collective branch of ShardedDPF._allreduce_ — device
    all_reduce / reduce_scatter_tensor + all_gather_into_tensor — which a
    gloo test can never reach.  On an 8-GPU node the identical code runs
    with world=8 over xGMI."""
    import torch.distributed as td

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    td.init_process_group("nccl", rank=0, world_size=1)
    try:
        from gpudpf import DPF, ShardedDPF, _core

        N = 1 << 14
        prf = DPF.PRF_CHACHA20
        torch.manual_seed(11)
        table = torch.randint(-(2**31), 2**31 - 1, (N, 16),
                              dtype=torch.int64).to(torch.int32)
        idxs = [0, 999, N - 1]
        ks = []
        for i in idxs:
            k1, k2 = _core.gen(i, N, b"nccl-%d" % i, prf)
            ks.append((torch.from_numpy(k1), torch.from_numpy(k2)))

        sd = ShardedDPF(prf=prf, device="cuda:0", collective=collective)
        sd.eval_init(table)

        # device-resident path: result must stay on the GPU
        a_dev = sd.eval_gpu([k[0] for k in ks], to_host=False)
        assert a_dev.is_cuda, "to_host=False must return a device tensor"
        b_dev = sd.eval_gpu([k[1] for k in ks], to_host=False)
        rec = (a_dev.to(torch.int64) - b_dev.to(torch.int64)).to(
            torch.int32).cpu().numpy()
        ok = bool(np.array_equal(rec, table[idxs, :].numpy()))

        # serving path: subkeys pre-sharded, eval_gpu_into + in-place RCCL
        subs1 = sd.shard_subkeys([k[0] for k in ks]).to("cuda:0").contiguous()
        subs2 = sd.shard_subkeys([k[1] for k in ks]).to("cuda:0").contiguous()
        o1 = torch.empty((len(idxs), 16), dtype=torch.int32, device="cuda:0")
        o2 = torch.empty_like(o1)
        sd.eval_gpu_into(subs1, o1)
        sd.eval_gpu_into(subs2, o2)
        rec2 = (o1.to(torch.int64) - o2.to(torch.int64)).to(
            torch.int32).cpu().numpy()
        ok = ok and bool(np.array_equal(rec2, table[idxs, :].numpy()))

        # sharded GraphedServer: graph-replayed local partial + RCCL
        # all-reduce outside the graph
        from gpudpf.serving import GraphedServer

        srv = GraphedServer(sd, batch=len(idxs))
        sa = srv.eval(torch.stack([k[0] for k in ks]))
        sb = srv.eval(torch.stack([k[1] for k in ks]))
        rec3 = (sa.to(torch.int64) - sb.to(torch.int64)).to(
            torch.int32).numpy()
        ok = ok and bool(np.array_equal(rec3, table[idxs, :].numpy()))
        q.put((0, ok, ""))
    except Exception as e:  # pragma: no cover
        q.put((0, False, repr(e)))
    finally:
        td.destroy_process_group()


@pytest.mark.parametrize("collective", ["all_reduce", "rs_ag"])
def test_sharded_rccl_branch_executes(collective):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_nccl_worker, args=(29797, collective, q))
    p.start()
    rank, ok, err = q.get(timeout=300)
    p.join(timeout=60)
    assert ok, f"rank {rank}: {err}"


def _nccl_wide_worker(port, q):
    """Sharded + wide entries (two-stage streaming-GEMM local path) under
    a real nccl process group."""
    import torch.distributed as td

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    td.init_process_group("nccl", rank=0, world_size=1)
    try:
        from gpudpf import DPF, ShardedDPF, _core

        N, e = 1 << 13, 40
        prf = DPF.PRF_SALSA20
        torch.manual_seed(5)
        table = torch.randint(-(2**31), 2**31 - 1, (N, e),
                              dtype=torch.int64).to(torch.int32)
        idxs = [1, 4095, 8000]
        sd = ShardedDPF(prf=prf, device="cuda:0")
        sd.eval_init(table)
        ks = [(_core.gen(i, N, b"w-%d" % i, prf)) for i in idxs]
        a = sd.eval_gpu([torch.from_numpy(k[0]) for k in ks], to_host=False)
        assert a.is_cuda and a.shape == (3, e)
        b = sd.eval_gpu([torch.from_numpy(k[1]) for k in ks], to_host=False)
        rec = (a.to(torch.int64) - b.to(torch.int64)).to(
            torch.int32).cpu().numpy()
        ok = bool(np.array_equal(rec, table[idxs, :].numpy()))
        q.put((0, ok, ""))
    except Exception as e:  # pragma: no cover
        q.put((0, False, repr(e)))
    finally:
        td.destroy_process_group()


def test_sharded_wide_entries_rccl():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_nccl_wide_worker, args=(29799, q))
    p.start()
    rank, ok, err = q.get(timeout=300)
    p.join(timeout=60)
    assert ok, f"rank {rank}: {err}"


def _nccl_stream_ingest_worker(port, q):
    """Sharded streaming ingest (eval_init_empty + global-index
    table_write): shard rows never exist on the host as a full table."""
    import torch.distributed as td

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    td.init_process_group("nccl", rank=0, world_size=1)
    try:
        from gpudpf import DPF, ShardedDPF, _core

        N, e = 1 << 13, 16
        prf = DPF.PRF_AES128
        torch.manual_seed(6)
        table = torch.randint(-(2**31), 2**31 - 1, (N, e),
                              dtype=torch.int64).to(torch.int32)
        sd = ShardedDPF(prf=prf, device="cuda:0")
        sd.eval_init_empty(N, e)
        for lo in range(0, N, 1000):  # global-index streaming fill
            hi = min(N, lo + 1000)
            sd.table_write(torch.arange(lo, hi), table[lo:hi])
        idxs = [2, 777, N - 1]
        back = sd.table_read(torch.tensor(
            [i for i in idxs if i % sd.world == sd.rank]))
        ok = back.shape[1] == e
        ks = [(_core.gen(i, N, b"si-%d" % i, prf)) for i in idxs]
        a = sd.eval_gpu([torch.from_numpy(k[0]) for k in ks])
        b = sd.eval_gpu([torch.from_numpy(k[1]) for k in ks])
        rec = (a.to(torch.int64) - b.to(torch.int64)).to(
            torch.int32).numpy()
        ok = ok and bool(np.array_equal(rec, table[idxs, :].numpy()))
        q.put((0, ok, ""))
    except Exception as exc:  # pragma: no cover
        q.put((0, False, repr(exc)))
    finally:
        td.destroy_process_group()


def test_sharded_streaming_ingest_rccl():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_nccl_stream_ingest_worker, args=(29801, q))
    p.start()
    rank, ok, err = q.get(timeout=300)
    p.join(timeout=60)
    assert ok, f"rank {rank}: {err}"
