"""One-off GPU stress validation (not a pytest file): roundtrips across
sizes x PRFs, concurrent DPF instances, repeated evals."""
import random
import sys

import numpy as np
import torch

sys.path.insert(0, "/root/repo")
from gpudpf import DPF


def roundtrip(n, batch, entrysize, prf):
    dpf = DPF(prf=prf)
    k1s, k2s, gt = [], [], []
    for _ in range(batch):
        i = random.randint(0, n - 1)
        gt.append(i)
        a, b = dpf.gen(i, n)
        k1s.append(a)
        k2s.append(b)
    table = torch.randint(-(2**31), 2**31 - 1, (n, entrysize), dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)
    ra, rb = dpf.eval_gpu(k1s), dpf.eval_gpu(k2s)
    rec = (ra.to(torch.int64) - rb.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[gt, :]), (n, batch, entrysize, prf)
    return dpf


random.seed(999)
for prf in [DPF.PRF_DUMMY, DPF.PRF_SALSA20, DPF.PRF_CHACHA20, DPF.PRF_AES128]:
    for n in [1 << 14, 1 << 16, 1 << 18, 1 << 20]:
        roundtrip(n, 24, random.randint(1, 16), prf)
        print(f"ok prf={prf} n={n}")

# concurrent instances on one device, interleaved evals
d1 = roundtrip(1 << 16, 8, 16, DPF.PRF_AES128)
d2 = roundtrip(1 << 18, 8, 16, DPF.PRF_SALSA20)
k1, k2 = d1.gen(7, 1 << 16)
j1, j2 = d2.gen(9, 1 << 18)
for _ in range(5):
    a = d1.eval_gpu([k1]); b = d2.eval_gpu([j1])
    a2 = d1.eval_gpu([k2]); b2 = d2.eval_gpu([j2])
    r1 = (a.to(torch.int64) - a2.to(torch.int64)).to(torch.int32)
    r2 = (b.to(torch.int64) - b2.to(torch.int64)).to(torch.int32)
    assert torch.equal(r1[0], d1.table[7]) and torch.equal(r2[0], d2.table[9])
print("interleaved instances ok")
print("STRESS PASS")
