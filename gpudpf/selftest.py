"""Install self-check (the reference's `python dpf.py` test battery,
dpf.py:139-367, as a module): run `python -m gpudpf.selftest`.

Runs the six reference checks: CPU one-hot, CPU table, GPU end-to-end,
GPU nopad, GPU sweep, GPU perf (GPU checks are skipped without a GPU).
"""

import random
import time

import numpy as np
import torch

from gpudpf import DPF


def test_cpu_dpf_one_hot(N=1024):
    dpf = DPF()
    K = 42
    k1, k2 = dpf.gen(K, N)
    v1 = dpf.eval_cpu([k1], one_hot_only=True)
    v2 = dpf.eval_cpu([k2], one_hot_only=True)
    rec = (v1 - v2).numpy()
    gt = np.zeros(rec.shape)
    gt[:, K] = 1
    assert np.linalg.norm(rec - gt) <= 1e-8
    print("Pass CPU (one-hot only) check.")


def _keys_and_table(dpf, N, batch, entrysize):
    k1s, k2s, gt_indices = [], [], []
    for _ in range(batch):
        indx = random.randint(0, N - 1)
        gt_indices.append(indx)
        k1, k2 = dpf.gen(indx, N)
        k1s.append(k1)
        k2s.append(k2)
    table = torch.randint(-(2**31), 2**31 - 1, (N, entrysize),
                          dtype=torch.int64).to(torch.int32)
    return k1s, k2s, gt_indices, table


def test_cpu_dpf(N=1024):
    dpf = DPF(device="cpu")
    k1s, k2s, gt, table = _keys_and_table(dpf, N, 64, 16)
    dpf.eval_init(table)
    rec = (dpf.eval_cpu(k1s).to(torch.int64) -
           dpf.eval_cpu(k2s).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[gt, :])
    print("Pass CPU check.")


def test_gpu_dpf(N=8192, batch=64, entrysize=16):
    dpf = DPF()
    k1s, k2s, gt, table = _keys_and_table(dpf, N, batch, entrysize)
    dpf.eval_init(table)
    rec = (dpf.eval_gpu(k1s).to(torch.int64) -
           dpf.eval_gpu(k2s).to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[gt, :])
    print("Pass GPU check.")


def test_gpu_dpf_nopad(N=8192, batch=42, entrysize=13):
    test_gpu_dpf(N, batch, entrysize)
    print("Pass GPU (nopad) check.")


def test_gpu_dpf_sweep():
    for n in [128, 256, 512, 1024, 8192]:
        test_gpu_dpf(n, batch=random.randint(1, 2559),
                     entrysize=random.randint(1, 15))
    print("Pass GPU (sweep) check.")


def test_gpu_dpf_perf(N=2048, batch=512, entrysize=16, prf=None):
    dpf = DPF(prf=prf)
    k1, _ = dpf.gen(1, N)
    keys = torch.stack([k1] * batch)
    table = torch.randint(-(2**31), 2**31 - 1, (N, entrysize),
                          dtype=torch.int64).to(torch.int32)
    dpf.eval_init(table)
    t0 = time.time()
    while time.time() - t0 < 0.5:
        dpf.eval_gpu(keys)
    torch.cuda.synchronize()
    tstart = time.time()
    reps = 10
    for _ in range(reps):
        dpf.eval_gpu(keys)
    elapsed = time.time() - tstart
    print("%s Key Size: %d bytes, Perf: %d dpfs/sec"
          % (dpf, int(k1.numel()) * 4, batch * reps / elapsed))


if __name__ == "__main__":
    random.seed(time.time())
    test_cpu_dpf()
    test_cpu_dpf_one_hot()
    if torch.cuda.is_available():
        test_gpu_dpf()
        test_gpu_dpf_nopad()
        test_gpu_dpf_sweep()
        test_gpu_dpf_perf()
    else:
        print("(no GPU: GPU checks skipped)")
