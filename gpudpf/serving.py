"""Production serving helpers.

GraphedServer: a hipGraph-captured serving step for fixed-batch key
streams — the launch-bound inner loop is captured once with
torch.cuda.CUDAGraph (hipGraph on ROCm) and replayed per batch, leaving
only the key H2D copy and result D2H on the host path.  This is the
idiomatic MI355X replacement for the reference's per-call stream
creation (dpf_wrapper.cu:155-156).

Serves all backend shapes:
  * fused (entry <= 16 words): one captured zero+fused launch;
  * wide entries: captured expand + streaming-GEMM sequence (the MFMA
    digit-plane path allocates per call and is not graph-capturable;
    the streaming GEMM reads the table in place, so the whole two-stage
    pipeline replays from one graph);
  * ShardedDPF: the local fused partial is graph-replayed and the RCCL
    all-reduce runs outside the graph (collectives are not captured).

Measured at batch=512 the fused graph is throughput-NEUTRAL vs plain
launches (0.58 vs 0.55 ms at n=16384): ROCm launch overhead for a single
fused kernel is already small.  The value is the pinned, preallocated,
fixed-address serving loop, and launch-count amortization for the
multi-launch wide path.
"""

import torch

from gpudpf.dpf import DPF, _hip


class GraphedServer:
    """Fixed-batch PIR serving loop with hipGraph replay.

    Usage:
        srv = GraphedServer(dpf, batch=512)   # dpf already eval_init'ed
        shares = srv.eval(keys_cpu)           # [batch, e] int32 (CPU)

    `dpf` may be a DPF or a ShardedDPF (then keys passed to eval must be
    full-domain keys; the subkey restriction runs on the host and the
    partial-sum all-reduce runs after graph replay).
    """

    MAX_WIDE_SHARES_BYTES = 16 << 30

    def __init__(self, dpf, batch: int):
        self.sharded = None
        if hasattr(dpf, "local"):  # ShardedDPF
            self.sharded = dpf
            dpf = dpf.local
        if dpf._table_gpu is None:
            raise Exception("eval_init the DPF before building a server")
        self.dpf = dpf
        self.batch = batch
        dev = dpf._table_gpu.device
        self.device = dev
        self.wide = dpf._entry_padded > DPF.ENTRY_SIZE
        if self.sharded is not None and self.wide:
            raise Exception("sharded GraphedServer serves the fused path "
                            "(entry <= 16 words)")
        self._keys_gpu = torch.zeros((batch, DPF.KEY_INTS), dtype=torch.int32,
                                     device=dev)
        self._keys_pinned = torch.zeros((batch, DPF.KEY_INTS),
                                        dtype=torch.int32).pin_memory()
        n = dpf._n_domain
        ep = dpf._entry_padded
        self._out_gpu = torch.zeros((batch, ep), dtype=torch.int32,
                                    device=dev)
        if self.wide:
            shares_bytes = batch * n * 4
            if shares_bytes > self.MAX_WIDE_SHARES_BYTES:
                raise Exception(
                    "wide GraphedServer share buffer would need %d bytes "
                    "(> %d); lower the batch" %
                    (shares_bytes, self.MAX_WIDE_SHARES_BYTES))
            self._shares_gpu = torch.zeros((batch, n), dtype=torch.int32,
                                           device=dev)

        def launch():
            stream = torch.cuda.current_stream(dev).cuda_stream
            self._out_gpu.zero_()
            if self.wide:
                _hip.eval_expand(self._keys_gpu.data_ptr(),
                                 self._shares_gpu.data_ptr(), dpf._aes_ptr,
                                 batch, n, dpf._depth, dpf._zlog,
                                 dpf.prf_method, stream)
                for lo in range(0, batch, 16):
                    hi = min(batch, lo + 16)
                    _hip.gemm_u32_stream(
                        self._shares_gpu[lo:hi].data_ptr(),
                        dpf._table_gpu.data_ptr(),
                        self._out_gpu[lo:hi].data_ptr(), hi - lo, n, ep,
                        stream)
            else:
                _hip.eval_fused(self._keys_gpu.data_ptr(),
                                dpf._table_gpu.data_ptr(),
                                self._out_gpu.data_ptr(), dpf._aes_ptr,
                                batch, n, dpf._depth, dpf._zlog,
                                dpf.prf_method, stream)

        # warm up on a side stream, then capture
        s = torch.cuda.Stream(dev)
        s.wait_stream(torch.cuda.current_stream(dev))
        with torch.cuda.stream(s):
            for _ in range(3):
                launch()
        torch.cuda.current_stream(dev).wait_stream(s)
        torch.cuda.synchronize(dev)
        self._graph = torch.cuda.CUDAGraph()
        # thread_local: the default 'global' capture mode errors ANY other
        # thread's CUDA call during capture — under a nccl process group
        # the RCCL watchdog thread races that window
        with torch.cuda.graph(self._graph, capture_error_mode="thread_local"):
            launch()

    def eval(self, keys, to_host=True):
        """keys: [batch, 524] int32 CPU tensor (or list) — full-domain
        keys for a sharded server, which restricts them per rank.
        Returns [batch, e] shares (CPU by default)."""
        if not isinstance(keys, torch.Tensor):
            keys = torch.stack([k.reshape(-1) for k in keys])
        if self.sharded is not None:
            keys = self.sharded.shard_subkeys(keys)
        if keys.shape[0] != self.batch:
            raise Exception("GraphedServer is fixed at batch=%d" % self.batch)
        # raw memcpy into the pinned staging buffer: torch's copy_ into a
        # pinned tensor device-synchronizes when an async H2D from it is
        # still pending (measured 5+ ms); numpy assignment does not.
        self._keys_pinned.numpy()[:] = keys.numpy()
        self._keys_gpu.copy_(self._keys_pinned, non_blocking=True)
        self._graph.replay()
        out = self._out_gpu
        if self.sharded is not None:
            out = self.sharded._allreduce_(out)
        out = out[:, : self.dpf.table_effective_entry_size]
        return out.cpu() if to_host else out


class PipelinedServer:
    """Double-buffered throughput serving: batch i+1's host staging and
    H2D copy overlap batch i's kernel (the reference's own benchmark
    interleaves iterations on two CUDA streams the same way,
    dpf_benchmark.cu:191-231).  At small n the fused kernel is ~0.15 ms
    while key staging costs ~0.25 ms of host/copy time — the pipeline
    hides it.

    Usage (throughput loop):
        srv = PipelinedServer(dpf, batch=512)     # dpf eval_init'ed
        h = srv.submit(keys_cpu)                  # non-blocking
        shares = srv.collect(h)                   # [batch, e] int32 CPU
    Submitting more than `depth` batches blocks until a slot drains.
    Each batch still performs the full step (key H2D, fused kernel,
    share D2H) — only ADJACENT batches overlap.
    """

    def __init__(self, dpf: DPF, batch: int, depth: int = 2):
        if dpf._table_gpu is None:
            raise Exception("eval_init the DPF before building a server")
        if dpf._entry_padded != DPF.ENTRY_SIZE:
            raise Exception("PipelinedServer serves the fused path "
                            "(entry <= 16 words)")
        self.dpf = dpf
        self.batch = batch
        self.depth = depth
        dev = dpf._table_gpu.device
        self.device = dev
        n = dpf._n_domain
        self._slots = []
        for _ in range(depth):
            slot = {
                "stream": torch.cuda.Stream(dev),
                "keys_pinned": torch.zeros((batch, DPF.KEY_INTS),
                                           dtype=torch.int32).pin_memory(),
                "keys_gpu": torch.zeros((batch, DPF.KEY_INTS),
                                        dtype=torch.int32, device=dev),
                "out_gpu": torch.zeros((batch, DPF.ENTRY_SIZE),
                                       dtype=torch.int32, device=dev),
                "out_pinned": torch.zeros((batch, DPF.ENTRY_SIZE),
                                          dtype=torch.int32).pin_memory(),
                "event": torch.cuda.Event(),
                "busy": False,
            }

            def launch(s=slot):
                stream = torch.cuda.current_stream(dev).cuda_stream
                s["out_gpu"].zero_()
                _hip.eval_fused(s["keys_gpu"].data_ptr(),
                                dpf._table_gpu.data_ptr(),
                                s["out_gpu"].data_ptr(), dpf._aes_ptr, batch,
                                n, dpf._depth, dpf._zlog, dpf.prf_method,
                                stream)

            warm = torch.cuda.Stream(dev)
            warm.wait_stream(torch.cuda.current_stream(dev))
            with torch.cuda.stream(warm):
                for _ in range(3):
                    launch()
            torch.cuda.current_stream(dev).wait_stream(warm)
            torch.cuda.synchronize(dev)
            slot["graph"] = torch.cuda.CUDAGraph()
            with torch.cuda.graph(slot["graph"], stream=slot["stream"],
                                  capture_error_mode="thread_local"):
                launch()
            self._slots.append(slot)
        self._next = 0

    def submit(self, keys):
        """Stage + launch a batch on the next slot (blocks only if the
        slot's previous batch has not been collected)."""
        if not isinstance(keys, torch.Tensor):
            keys = torch.stack([k.reshape(-1) for k in keys])
        if keys.shape[0] != self.batch:
            raise Exception("PipelinedServer is fixed at batch=%d"
                            % self.batch)
        slot = self._slots[self._next]
        self._next = (self._next + 1) % self.depth
        if slot["busy"]:
            slot["event"].synchronize()  # previous user never collected
            slot["busy"] = False
        slot["keys_pinned"].numpy()[:] = keys.numpy()
        with torch.cuda.stream(slot["stream"]):
            slot["keys_gpu"].copy_(slot["keys_pinned"], non_blocking=True)
            slot["graph"].replay()
            slot["out_pinned"].copy_(slot["out_gpu"], non_blocking=True)
            slot["event"].record(slot["stream"])
        slot["busy"] = True
        return slot

    def collect(self, slot):
        """Wait for a submitted batch and return its [batch, e] shares."""
        slot["event"].synchronize()
        slot["busy"] = False
        return slot["out_pinned"][:, : self.dpf.table_effective_entry_size
                                  ].clone()


class TwoStageServer:
    """Pipelined wide-entry serving (entry > 16 words): batch i+1's
    one-hot expansion runs on its own stream while batch i's streaming
    GEMM reads the table — steady-state step = max(expansion, GEMM)
    instead of their sum.  On a 240 GB table (2^28 x 224) the serial
    two-stage step is ~53 ms expansion + ~64 ms GEMM; pipelined serving
    approaches the larger of the two.

    Usage:
        srv = TwoStageServer(dpf, batch=8)   # dpf eval_init'ed, wide
        h = srv.submit(keys_cpu)             # non-blocking
        shares = srv.collect(h)              # [batch, e] int32 CPU
    """

    def __init__(self, dpf: DPF, batch: int, depth: int = 2):
        if dpf._table_gpu is None:
            raise Exception("eval_init the DPF before building a server")
        if dpf._entry_padded <= DPF.ENTRY_SIZE:
            raise Exception("TwoStageServer serves wide entries "
                            "(> 16 words); use PipelinedServer")
        if batch > 16:
            raise Exception("streaming-GEMM batch is capped at 16 per "
                            "table pass; lower the batch")
        self.dpf = dpf
        self.batch = batch
        self.depth = depth
        dev = dpf._table_gpu.device
        self.device = dev
        n = dpf._n_domain
        ep = dpf._entry_padded
        free, _total = torch.cuda.mem_get_info(dev)
        need = depth * batch * n * 4
        if need > free - (2 << 30):
            raise Exception("pipeline share buffers need %d bytes "
                            "(free %d); lower batch/depth" % (need, free))
        self._gemm_stream = torch.cuda.Stream(dev)
        self._slots = []
        for _ in range(depth):
            self._slots.append({
                "stream": torch.cuda.Stream(dev),
                "keys_pinned": torch.zeros((batch, DPF.KEY_INTS),
                                           dtype=torch.int32).pin_memory(),
                "keys_gpu": torch.zeros((batch, DPF.KEY_INTS),
                                        dtype=torch.int32, device=dev),
                "shares": torch.zeros((batch, n), dtype=torch.int32,
                                      device=dev),
                "out_gpu": torch.zeros((batch, ep), dtype=torch.int32,
                                       device=dev),
                "ev_expand": torch.cuda.Event(),
                "ev_done": torch.cuda.Event(),
                "busy": False,
            })
        self._next = 0

    def submit(self, keys):
        if not isinstance(keys, torch.Tensor):
            keys = torch.stack([k.reshape(-1) for k in keys])
        if keys.shape[0] != self.batch:
            raise Exception("TwoStageServer is fixed at batch=%d"
                            % self.batch)
        dpf = self.dpf
        slot = self._slots[self._next]
        self._next = (self._next + 1) % self.depth
        if slot["busy"]:
            slot["ev_done"].synchronize()
            slot["busy"] = False
        slot["keys_pinned"].numpy()[:] = keys.numpy()
        n = dpf._n_domain
        with torch.cuda.stream(slot["stream"]):
            slot["keys_gpu"].copy_(slot["keys_pinned"], non_blocking=True)
            _hip.eval_expand(slot["keys_gpu"].data_ptr(),
                             slot["shares"].data_ptr(), dpf._aes_ptr,
                             self.batch, n, dpf._depth, dpf._zlog,
                             dpf.prf_method, slot["stream"].cuda_stream)
            slot["ev_expand"].record(slot["stream"])
        # the GEMM stream serializes table passes across slots (the table
        # stream is the contended resource); it only waits for THIS
        # slot's expansion
        self._gemm_stream.wait_event(slot["ev_expand"])
        with torch.cuda.stream(self._gemm_stream):
            slot["out_gpu"].zero_()
            _hip.gemm_u32_stream(slot["shares"].data_ptr(),
                                 dpf._table_gpu.data_ptr(),
                                 slot["out_gpu"].data_ptr(), self.batch, n,
                                 dpf._entry_padded,
                                 self._gemm_stream.cuda_stream)
            slot["ev_done"].record(self._gemm_stream)
        slot["busy"] = True
        return slot

    def collect(self, slot):
        slot["ev_done"].synchronize()
        slot["busy"] = False
        return slot["out_gpu"][:, : self.dpf.table_effective_entry_size
                               ].cpu()
