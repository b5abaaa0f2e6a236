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

The GPU hot path is device-resident end to end: keys go host->device once,
the fused kernel writes partials into a device buffer, the RCCL collective
reduces that buffer in place over xGMI, and the result stays on device
(`to_host=False`) for serving pipelines — zero host round-trips per batch.

Also provided: ReplicatedDPF — each rank holds the full table and serves
its own key stream (the weak-scaling / production throughput mode).
"""

import warnings

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

    def eval_init_empty(self, n, e):
        """Streaming-ingest mode for aggregate tables near W x HBM: each
        rank allocates its n/W-row shard directly in device memory (never
        on the host) and fills it with table_write."""
        n = int(n)
        if n & (n - 1) != 0:
            raise Exception("sharded tables must have power-of-two entries")
        if n % self.world != 0:
            raise Exception("n must be divisible by world size")
        self.table_num_entries = n
        self.table_effective_entry_size = int(e)
        self.local.eval_init_empty(n // self.world, e)

    def table_write(self, indices, rows):
        """Write natural GLOBAL-index rows; each rank keeps its residue
        class (idx %% W == rank) and ignores the rest, so every rank can
        be fed the same stream."""
        idx = indices if isinstance(indices, torch.Tensor) \
            else torch.as_tensor(indices, dtype=torch.int64)
        mine = (idx % self.world) == self.rank
        if not bool(mine.any()):
            return
        self.local.table_write(idx[mine] // self.world, rows[mine])

    def table_read(self, indices):
        """Gather natural global rows owned by THIS rank (indices must
        all satisfy idx %% W == rank)."""
        idx = indices if isinstance(indices, torch.Tensor) \
            else torch.as_tensor(indices, dtype=torch.int64)
        if bool(((idx % self.world) != self.rank).any()):
            raise Exception("table_read: some indices belong to other ranks")
        return self.local.table_read(idx // self.world)

    def shard_subkeys(self, keys):
        """Restrict a batch of full-domain wire-format keys to this rank's
        residue class: [b, 524] int32 CPU tensor of depth-log2(W) subkeys.
        Host-side (the restriction walks log2(W) PRF levels per key), one
        C++ call for the whole batch; on a serving path do this once per
        batch, then feed eval_gpu_into."""
        if not isinstance(keys, torch.Tensor):
            keys = torch.stack([k.reshape(-1) for k in keys])
        if keys.dim() == 1:
            keys = keys.unsqueeze(0)
        return torch.from_numpy(
            _core.shard_subkey_batch(keys.contiguous().numpy(),
                                     self.prf_method, self.rank, self.world))

    # backwards-compatible internal alias
    _subkeys = shard_subkeys

    def eval_gpu(self, keys, to_host=True):
        """Evaluate a batch of full-domain keys cooperatively.  Partials
        stay on device: fused kernel -> device buffer -> in-place RCCL
        all-reduce over xGMI.  Returns [b, e] int32 shares — a CPU tensor
        by default, or the device tensor when to_host=False (the serving
        hot path does zero host round-trips)."""
        subs = self.shard_subkeys(keys)
        if self.local._entry_padded == DPF.ENTRY_SIZE:
            # fused path, fully device-resident
            dev = self.local._table_gpu.device
            keys_gpu = subs.to(dev, non_blocking=True).contiguous()
            part = torch.empty((subs.shape[0], DPF.ENTRY_SIZE),
                               dtype=torch.int32, device=dev)
            self.local.eval_gpu_into(keys_gpu, part)
            part = self._allreduce_(part)
            part = part[:, : self.table_effective_entry_size]
        else:
            # wide entries: two-stage GEMM path (device-resident output);
            # the trimmed [b, e] view of the padded output is not
            # contiguous, and RCCL collectives need dense buffers
            part = self.local.eval_gpu(subs, out_device=True).contiguous()
            part = self._allreduce_(part)
        return part.cpu() if to_host else part

    def eval_gpu_into(self, subkeys_gpu, out_gpu):
        """Serving path: subkeys (from shard_subkeys) already on device,
        fused partials written into out_gpu [b,16] and all-reduced in
        place.  No host traffic at all."""
        self.local.eval_gpu_into(subkeys_gpu, out_gpu)
        self._allreduce_(out_gpu)

    def eval_cpu(self, keys):
        part = self.local.eval_cpu(self.shard_subkeys(keys))
        return self._allreduce_(part.contiguous())

    def _allreduce_(self, buf):
        """In-place int32 sum reduction of `buf` (wraparound == exact
        mod-2^32).  For the nccl(=RCCL) backend buf must already live on
        this rank's GPU — it is reduced in place with no host copies."""
        backend = td.get_backend(self.group)
        if backend == "nccl":
            if not buf.is_cuda:  # eval_cpu under an RCCL process group
                dev = torch.device(self.local.device or "cuda")
                gbuf = buf.to(dev)
                td.all_reduce(gbuf, op=td.ReduceOp.SUM, group=self.group)
                buf.copy_(gbuf.cpu())
                return buf
            if self.collective == "rs_ag":
                if buf.shape[0] % self.world != 0:
                    warnings.warn(
                        "collective='rs_ag' needs batch %% world == 0 "
                        "(batch=%d, world=%d); falling back to all_reduce "
                        "for this batch" % (buf.shape[0], self.world))
                    td.all_reduce(buf, op=td.ReduceOp.SUM, group=self.group)
                else:
                    chunk = torch.empty_like(buf[: buf.shape[0] // self.world])
                    td.reduce_scatter_tensor(chunk, buf, op=td.ReduceOp.SUM,
                                             group=self.group)
                    td.all_gather_into_tensor(buf, chunk, group=self.group)
            else:
                td.all_reduce(buf, op=td.ReduceOp.SUM, group=self.group)
            return buf
        # gloo (CPU rendezvous / tests): collective runs on host memory
        host = buf.cpu() if buf.is_cuda else buf
        td.all_reduce(host, op=td.ReduceOp.SUM, group=self.group)
        if buf.is_cuda:
            buf.copy_(host)
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
