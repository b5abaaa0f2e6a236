"""gpudpf — MI355X-native Distributed Point Function / 2-server PIR engine.

A from-scratch AMD CDNA4 framework with the capabilities of
facebookresearch/GPU-DPF: CPU key generation, hand-written HIP (gfx950)
batched DPF evaluation with a fused table inner product, multi-GPU row
sharding over RCCL/xGMI, and the batch-PIR co-design toolkit.

Public API parity (reference dpf.py:35-137):

    from gpudpf import DPF
    d = DPF(prf=DPF.PRF_AES128)
    k1, k2 = d.gen(k, n)
    d.eval_init(table)          # [n, e] int32 (any n; e<=16 uses the fused path)
    shares = d.eval_gpu([k1])   # [1, e] int32 secret shares
"""

try:
    from gpudpf.dpf import DPF  # noqa: F401
    from gpudpf.dist import ReplicatedDPF, ShardedDPF  # noqa: F401
    from gpudpf.serving import GraphedServer  # noqa: F401
except ImportError:
    # Fresh checkout: the native extensions are not built yet.  Importing
    # the bare package must still work so `gpudpf._build` can bootstrap
    # (`python -m gpudpf._build` or __graft_entry__.build()).
    DPF = None  # type: ignore[assignment]

__version__ = "0.1.0"
