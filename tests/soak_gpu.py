"""GPU soak: continuous mixed-config evaluation with reconstruction checks
(run standalone; duration via argv[1] seconds, default 240)."""
import random
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from gpudpf import DPF

DUR = float(sys.argv[1]) if len(sys.argv) > 1 else 240.0
random.seed(31337)
engines = {}
checked = 0
t0 = time.time()
while time.time() - t0 < DUR:
    prf = random.choice([DPF.PRF_SALSA20, DPF.PRF_CHACHA20, DPF.PRF_AES128])
    n = 1 << random.choice([14, 16, 17, 18, 20])
    key = (prf, n)
    if key not in engines:
        d = DPF(prf=prf)
        table = torch.randint(-(2**31), 2**31 - 1, (n, 16),
                              dtype=torch.int64).to(torch.int32)
        d.eval_init(table)
        engines[key] = d
    d = engines[key]
    batch = random.choice([1, 7, 64, 512])
    idxs = [random.randrange(n) for _ in range(batch)]
    k1s, k2s = d.gen_batch(idxs, n)
    rec = (d.eval_gpu(k1s).to(torch.int64) -
           d.eval_gpu(k2s).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, d.table[idxs, :]), (prf, n, batch)
    checked += batch
print(f"SOAK PASS: {checked} keys reconstructed across {len(engines)} configs "
      f"in {time.time()-t0:.0f}s")
