"""Flagship benchmark: batched DPF evaluation + fused PIR lookup (DPFs/sec).

Measures the reference's headline metric (DPFs/sec at
batch=512, entry=16 x u32) on MI355X.  Default config is the V100-baseline
flagship row: entries=2^20, AES-128 (V100: 923 DPFs/sec).

One "step" = one full server iteration for a 512-key batch per GPU: upload
the 512 wire-format keys (H2D), run the fused expand+dot kernel, download
the [512,16] shares (D2H) — the same work the reference times in
test_gpu_dpf_perf (dpf.py:286-320).

Multi-GPU (launched by torch.distributed.run, one rank per GPU): weak
scaling — every rank serves its own independent 512-key stream against a
replicated table; `value` is the aggregate DPFs/sec over all ranks, using
the MAX step time across ranks.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
       [--entries N] [--prf AES128|SALSA20|CHACHA20] [--batch B]
       [--mode replicate|shard]
"""

import argparse
import json
import os
import time

import torch

from gpudpf import DPF, ShardedDPF, _core

V100_BASELINE = {
    # AES-128, entry=16xu32, batch=512
    ("AES128", 16384): 52536.0,
    ("AES128", 65536): 15392.0,
    ("AES128", 262144): 3967.0,
    ("AES128", 1048576): 923.0,
    ("SALSA20", 16384): 145646.0,
    ("SALSA20", 65536): 54892.0,
    ("SALSA20", 262144): 16650.0,
    ("SALSA20", 1048576): 3894.0,
    ("CHACHA20", 16384): 139590.0,
    ("CHACHA20", 65536): 56120.0,
    ("CHACHA20", 262144): 16086.0,
    ("CHACHA20", 1048576): 4054.0,
}

PRF_IDS = {
    "DUMMY": DPF.PRF_DUMMY,
    "SALSA20": DPF.PRF_SALSA20,
    "CHACHA20": DPF.PRF_CHACHA20,
    "AES128": DPF.PRF_AES128,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--entries", type=int, default=1 << 20)
    ap.add_argument("--prf", type=str, default="AES128", choices=list(PRF_IDS))
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--entry-size", type=int, default=16)
    ap.add_argument("--mode", type=str, default="replicate",
                    choices=["replicate", "shard"])
    ap.add_argument("--device", type=str, default=None,
                    help="cuda (default when available) or cpu (CI runs of "
                         "the distributed path over gloo)")
    ap.add_argument("--backend", type=str, default=None,
                    choices=["nccl", "gloo"],
                    help="process-group backend (default: nccl on cuda, "
                         "gloo on cpu)")
    ap.add_argument("--check", action="store_true",
                    help="verify reconstruction correctness after timing "
                         "(adds a second key batch eval)")
    ap.add_argument("--no-graph", action="store_true",
                    help="plain per-step launches instead of the hipGraph "
                         "serving loop")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    on_gpu = (args.device or "cuda") != "cpu" and torch.cuda.is_available()
    backend = args.backend or ("nccl" if on_gpu else "gloo")
    if distributed:
        torch.distributed.init_process_group(backend)
        if on_gpu:
            torch.cuda.set_device(local_rank)
    device = ("cuda:%d" % local_rank) if on_gpu else "cpu"
    prf = PRF_IDS[args.prf]
    n = args.entries

    # Synthetic workload: random-init table, random target indices.  The
    # table is rank-independent in shard mode (all ranks shard ONE table)
    # and per-rank in replicate mode (independent serving streams).
    tseed = 1000 if args.mode == "shard" else 1000 + rank
    torch.manual_seed(tseed)
    table = torch.randint(-(2**31), 2**31 - 1, (n, args.entry_size),
                          dtype=torch.int32)
    # Keys: in shard mode every rank receives the SAME client batch (the
    # world evaluates it cooperatively); in replicate mode each rank has
    # its own stream.
    kseed = 0 if args.mode == "shard" else rank
    g = torch.Generator().manual_seed(kseed)
    alphas = [int(torch.randint(0, n, (1,), generator=g).item())
              for _ in range(args.batch)]
    keys, keys2 = [], []
    for i, alpha in enumerate(alphas):
        k1, k2 = _core.gen(alpha, n, b"bench-%d-%d" % (kseed, i), prf)
        keys.append(torch.from_numpy(k1))
        keys2.append(torch.from_numpy(k2))
    keys_cpu = torch.stack(keys).contiguous()
    keys2_cpu = torch.stack(keys2).contiguous()

    if args.mode == "shard" and distributed:
        engine = ShardedDPF(prf=prf, device=device)
        engine.eval_init(table)
        if on_gpu:
            if args.no_graph:
                def step(k=keys_cpu):
                    return engine.eval_gpu(k)
            else:
                from gpudpf.serving import GraphedServer
                srv = GraphedServer(engine, args.batch)
                def step(k=keys_cpu):
                    return srv.eval(k)
        else:
            def step(k=keys_cpu):
                return engine.eval_cpu(k)
    else:
        engine = DPF(prf=prf, device=device)
        engine.eval_init(table)
        if on_gpu:
            if args.no_graph or args.entry_size > 16:
                # wide entries route through the two-stage path; plain
                # per-step eval (the pipeline classes serve fixed shapes)
                def step(k=keys_cpu):
                    return engine.eval_gpu(k)
                def drain():
                    pass
            else:
                # production serving loop: hipGraph replay, double-
                # buffered so batch i+1's key staging/H2D overlaps batch
                # i's kernel (the reference's dual-stream iteration
                # interleave, dpf_benchmark.cu:191-231).  Every batch
                # still does the full step: key H2D, kernel, share D2H.
                from gpudpf.serving import PipelinedServer
                srv = PipelinedServer(
                    engine, args.batch,
                    depth=int(os.environ.get("GPUDPF_PIPE_DEPTH", "2")))
                pending = []
                def step(k=keys_cpu):
                    pending.append(srv.submit(k))
                    if len(pending) >= 2:
                        return srv.collect(pending.pop(0))
                def drain():
                    while pending:
                        srv.collect(pending.pop(0))
        else:
            def step(k=keys_cpu):
                return engine.eval_cpu(k)

    if args.mode == "shard" or not on_gpu or args.no_graph:
        def drain():  # noqa: F811 (no pipeline in these modes)
            pass

    def sync():
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    drain()
    sync()
    if distributed:
        torch.distributed.barrier()
        sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    drain()
    sync()
    t1 = time.perf_counter()
    if distributed:
        torch.distributed.barrier()
        sync()

    if args.check:
        if on_gpu:
            ea = engine.eval_gpu(keys_cpu)
            eb = engine.eval_gpu(keys2_cpu)
        else:
            ea = step(keys_cpu)
            eb = step(keys2_cpu)
        rec = (ea.to(torch.int64) - eb.to(torch.int64)).to(torch.int32)
        want = table[alphas, : args.entry_size]
        if not torch.equal(rec.cpu(), want):
            raise SystemExit("bench --check FAILED: reconstruction mismatch "
                             "(rank %d, mode %s)" % (rank, args.mode))

    elapsed = t1 - t0
    if distributed:
        buf = torch.tensor([elapsed], dtype=torch.float64)
        if torch.distributed.get_backend() == "nccl":
            buf = buf.to(device)
        torch.distributed.all_reduce(buf, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(buf.item())

    ms_per_step = elapsed / args.steps * 1000.0
    # aggregate: in replicate mode every rank serves its own batch; in shard
    # mode the whole world serves one batch cooperatively.
    batches_per_step = world if args.mode == "replicate" else 1
    value = args.batch * batches_per_step * args.steps / elapsed
    # published reference rows are entry_size=16 only
    base = V100_BASELINE.get((args.prf, n)) if args.entry_size == 16 else None

    if rank == 0:
        print(json.dumps({
            "metric": "DPFs/sec",
            "value": round(value, 1),
            "unit": "dpfs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak" if args.mode == "replicate" else "strong",
            "vs_baseline": round(value / base, 2) if base else None,
            "dtype": "int32",
            "data": "synthetic",
            "config": {
                "model": "dpf-pir",
                "entries": n,
                "entry_size": args.entry_size,
                "prf": args.prf,
                "batch_per_gpu": args.batch if args.mode == "replicate" else None,
                "global_batch": args.batch * batches_per_step,
                "parallelism": ("dp%d-replicated" % world) if args.mode == "replicate"
                               else ("shard%d-%s" % (world,
                                     "rccl" if backend == "nccl" else backend)),
            },
        }), flush=True)

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
