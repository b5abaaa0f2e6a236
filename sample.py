"""2-server PIR walkthrough (parity with the reference's sample.py).

A client wants entry `alpha` of a replicated table without revealing
`alpha` to either server.  It generates two DPF keys; each server expands
its key against the table on its MI355X and returns a [1,16] int32 share;
the client reconstructs entry = share_a - share_b.
"""

import torch

from gpudpf import DPF

N = 65536            # table entries (power of two)
ENTRY_SIZE = 16      # ints per entry
ALPHA = 4242         # the secret index


def server(dpf, key):
    # Each (untrusting) server sees only an opaque 2096-byte key.
    return dpf.eval_gpu([key]) if torch.cuda.is_available() else dpf.eval_cpu([key])


def main():
    # Server-side setup: both servers hold the same table.
    table = torch.arange(N * ENTRY_SIZE, dtype=torch.int64).reshape(N, ENTRY_SIZE)
    table = (table % (2**31)).to(torch.int32)

    dpf = DPF(prf=DPF.PRF_AES128)
    dpf.eval_init(table)

    # Client: generate the key pair for the secret index.
    k1, k2 = dpf.gen(ALPHA, N)
    print("key size: %d bytes each" % (k1.numel() * 4))

    # Servers: evaluate independently.
    a = server(dpf, k1)
    b = server(dpf, k2)

    # Client: reconstruct.
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
    expected = table[ALPHA]
    assert torch.equal(rec[0], expected), "reconstruction failed"
    print("recovered entry", ALPHA, ":", rec[0].tolist())
    print("PIR round-trip OK")


if __name__ == "__main__":
    main()
