"""Multi-GPU DPF evaluation over RCCL/xGMI (torch.distributed).

The reference has no distributed layer; this is the
MI355X-native extension: the table is row-sharded across ranks by residue
class (rank r owns natural rows idx % W == r), which restricts the GGM tree
to ONE subtree per rank — rank r walks log2(W) levels from the root
consuming the bits of r (these are the FIRST-consumed index bits), then
runs the normal fused kernel on its n/W-entry sub-domain.  Per-shard
partial inner products add elementwise mod 2^32 (u32 wraparound is exact
because truncation is a ring hom), so a single int32 sum all-reduce of the
[batch, 16] partials yields the full result on every rank.

Payloads are KB-scale (batch x 64 B), so the collective is latency-bound:
one fused all_reduce per batch, no bucketing needed.

Also provided: ReplicatedDPF — each rank holds the full table and serves
its own key stream (the weak-scaling / production throughput mode).
"""

import torch
import torch.distributed as td

from gpudpf import _core
from gpudpf.dpf import DPF


class ShardedDPF(object):
    """Strong-scaling mode: one batch of keys is evaluated cooperatively by
    all ranks against a row-sharded table."""

    def __init__(self, prf=None, device=None, group=None,
                 collective="all_reduce"):
        """collective: "all_reduce" (default — the [batch,16] payload is
        KB-scale and latency-bound, a fused all-reduce is optimal) or
        "rs_ag" (explicit reduce-scatter of per-shard partials +
        all-gather of results, NCCL/RCCL only — the decomposition a ring
        all-reduce performs internally, exposed for batch-sharded
        pipelines that consume only their own slice between the two
        phases)."""
        if not td.is_initialized():
            raise Exception("torch.distributed must be initialized")
        self.group = group
        self.world = td.get_world_size(group)
        self.rank = td.get_rank(group)
        if self.world & (self.world - 1) != 0:
            raise Exception("world size must be a power of two")
        self.prf_method = DPF.DEFAULT_PRF if prf is None else prf
        self.collective = collective
        self.local = DPF(prf=self.prf_method, device=device)
        self.table_num_entries = None
        self.table_effective_entry_size = None

    def shard_rows(self, table):
        """Rows of the full [n, e] table owned by this rank."""
        return table[self.rank :: self.world]

    def eval_init(self, table):
        """Initialize from the FULL table (each rank slices its residue
        class).  For tables too large to materialize, build the shard
        directly and call eval_init_local."""
        n = int(table.shape[0])
        if n & (n - 1) != 0:
            raise Exception("sharded tables must have power-of-two entries")
        self.table_num_entries = int(table.shape[0])
        self.table_effective_entry_size = int(table.shape[1])
        self.local.eval_init(self.shard_rows(table).contiguous())

    def eval_init_local(self, local_rows, n):
        """Initialize from this rank's shard only (local_rows[i] must be
        full-table row i*world + rank)."""
        self.table_num_entries = int(n)
        self.table_effective_entry_size = int(local_rows.shape[1])
        self.local.eval_init(local_rows.contiguous())

    def _subkeys(self, keys):
        if isinstance(keys, torch.Tensor) and keys.dim() == 2:
            keys = [keys[i] for i in range(keys.shape[0])]
        subs = [
            torch.from_numpy(
                _core.shard_subkey(
                    k.reshape(-1).numpy(), self.prf_method, self.rank, self.world
                )
            )
            for k in keys
        ]
        return torch.stack(subs)

    def eval_gpu(self, keys):
        part = self.local.eval_gpu(self._subkeys(keys))
        part = self._allreduce(part)
        return part

    def eval_cpu(self, keys):
        part = self.local.eval_cpu(self._subkeys(keys))
        part = self._allreduce(part)
        return part

    def _allreduce(self, part):
        # int32 sum with wraparound == exact mod-2^32 reduction
        backend = td.get_backend(self.group)
        if backend == "nccl":
            dev = torch.device(self.local.device)
            buf = part.to(dev).contiguous()
            if self.collective == "rs_ag" and buf.shape[0] % self.world == 0:
                chunk = torch.empty_like(buf[: buf.shape[0] // self.world])
                td.reduce_scatter_tensor(chunk, buf, op=td.ReduceOp.SUM,
                                         group=self.group)
                td.all_gather_into_tensor(buf, chunk, group=self.group)
            else:
                td.all_reduce(buf, op=td.ReduceOp.SUM, group=self.group)
            return buf.cpu()
        buf = part.clone()
        td.all_reduce(buf, op=td.ReduceOp.SUM, group=self.group)
        return buf


class ReplicatedDPF(object):
    """Weak-scaling mode: every rank holds the full table and evaluates its
    own key stream (production serving: aggregate throughput scales with
    ranks; no communication on the hot path)."""

    def __init__(self, prf=None, device=None):
        self.local = DPF(prf=prf, device=device)

    def eval_init(self, table):
        self.local.eval_init(table)

    def eval_gpu(self, keys, **kw):
        return self.local.eval_gpu(keys, **kw)
