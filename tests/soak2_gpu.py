import random, sys, time
import torch
sys.path.insert(0, "/root/repo")
from gpudpf import DPF

random.seed(777)
t0 = time.time()

# 1) real chunking: batch > MAX_LAUNCH_BATCH
n = 1 << 14
d = DPF(prf=DPF.PRF_SALSA20)
table = torch.randint(-(2**31), 2**31-1, (n,16), dtype=torch.int64).to(torch.int32)
d.eval_init(table)
idxs = [random.randrange(n) for _ in range(5000)]
k1s, k2s = d.gen_batch(idxs, n)
rec = (d.eval_gpu(k1s).to(torch.int64) - d.eval_gpu(k2s).to(torch.int64)).to(torch.int32)
assert torch.equal(rec, table[idxs, :]); print("chunked batch 5000 ok")

# 2) wide entries + non-pow2 mixed soak
for trial in range(12):
    n = random.choice([1000, 5000, 20000, 1 << 15])
    e = random.choice([3, 16, 24, 48])
    prf = random.choice([DPF.PRF_CHACHA20, DPF.PRF_AES128, DPF.PRF_DUMMY])
    d = DPF(prf=prf)
    table = torch.randint(-(2**31), 2**31-1, (n, e), dtype=torch.int64).to(torch.int32)
    d.eval_init(table)
    idxs = [random.randrange(n) for _ in range(48)]
    k1s, k2s = d.gen_batch(idxs, n)
    rec = (d.eval_gpu(k1s).to(torch.int64) - d.eval_gpu(k2s).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :]), (n, e, prf)
print("wide/non-pow2 soak ok")

# 3) two_stage strategy soak at entry<=16 (explicit)
for trial in range(6):
    n = 1 << random.choice([14, 16])
    d = DPF(prf=DPF.PRF_SALSA20)
    table = torch.randint(-(2**31), 2**31-1, (n,16), dtype=torch.int64).to(torch.int32)
    d.eval_init(table)
    idxs = [random.randrange(n) for _ in range(96)]
    k1s, k2s = d.gen_batch(idxs, n)
    a = d.eval_gpu(k1s, strategy="two_stage")
    b = d.eval_gpu(k2s, strategy="two_stage")
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :])
    assert torch.equal(a, d.eval_gpu(k1s))  # strategy equivalence
print("two_stage soak ok")
print("SOAK2 PASS in %.0fs" % (time.time()-t0))

# 5) round-2 paths: BFS strategy, graphs (fused + wide), streaming ingest
for trial in range(6):
    n = random.choice([512, 4096, 1 << 14])
    prf = random.choice([DPF.PRF_SALSA20, DPF.PRF_AES128])
    d = DPF(prf=prf)
    table = torch.zeros((n, 1), dtype=torch.int32)
    d.eval_init(table)
    idxs = [random.randrange(n) for _ in range(16)]
    k1s, k2s = d.gen_batch(idxs, n)
    oh = (d.eval_gpu(k1s, one_hot_only=True, strategy="bfs").to(torch.int64)
          - d.eval_gpu(k2s, one_hot_only=True, strategy="bfs").to(torch.int64))
    want = torch.zeros((16, n), dtype=torch.int64)
    for r, ix in enumerate(idxs):
        want[r, ix] = 1
    assert torch.equal(oh, want), ("bfs", n, prf)
print("bfs soak ok")

from gpudpf.serving import GraphedServer
for e, batch in ((16, 128), (40, 48)):
    n = 1 << 13
    d = DPF(prf=DPF.PRF_CHACHA20)
    table = torch.randint(-(2**31), 2**31-1, (n, e), dtype=torch.int64).to(torch.int32)
    d.eval_init(table)
    srv = GraphedServer(d, batch)
    for rep in range(10):
        idxs = [random.randrange(n) for _ in range(batch)]
        k1s, k2s = d.gen_batch(idxs, n)
        rec = (srv.eval(k1s).to(torch.int64) - srv.eval(k2s).to(torch.int64)).to(torch.int32)
        assert torch.equal(rec, table[idxs, :]), ("graph", e, rep)
print("graph replay soak ok (fused + wide)")

d = DPF(prf=DPF.PRF_SALSA20)
n, e = 1 << 14, 24
d.eval_init_empty(n, e)
ref = torch.zeros((n, e), dtype=torch.int32)
for rep in range(20):
    idx = torch.randint(0, n, (500,)).unique()
    rows = torch.randint(-(2**31), 2**31-1, (idx.numel(), e), dtype=torch.int64).to(torch.int32)
    d.table_write(idx, rows)
    ref[idx] = rows
probe = [random.randrange(n) for _ in range(24)]
k1s, k2s = d.gen_batch(probe, n)
rec = (d.eval_gpu(k1s).to(torch.int64) - d.eval_gpu(k2s).to(torch.int64)).to(torch.int32)
assert torch.equal(rec, ref[probe, :])
print("streaming-ingest soak ok")

print("soak2 total %.1fs" % (time.time() - t0))
