"""Batch-PIR optimization for on-device ML table access.

Re-implements (from scratch) the mechanics of the reference's
paper/experimental/batch_pir/batch_pir_optimization.py:

An application accesses a table of `n` embedding/vocab entries with a
batch of indices per inference.  Fetching every index through 2-server
PIR costs one DPF query per index; three co-design optimizations shrink
that cost at bounded accuracy impact:

  1. HOT/COLD SPLIT: the most frequently accessed entries (by training
     frequency) are replicated on-device ("hot"); only cold entries go
     through PIR.  Hot hits are free; the hot store costs device memory.
  2. COLLOCATION: entries that frequently co-occur in one batch are
     packed into the same (wider) table row, so one PIR fetch returns a
     group of likely-needed entries.
  3. BINNING (batch PIR): cold entries are hashed into `num_bins` bins
     (each bin is a smaller table); the client issues at most
     `queries_per_bin` PIR queries per bin per batch.  Indices that
     exceed a bin's query budget are NOT recovered that batch: the model
     must tolerate the miss (<unk> token / zero embedding), which is the
     accuracy cost this module measures.

Costs are accounted per batch:
  communication = num_queries x (2 x key_size + response)  bytes
  computation   = num_queries x (bin_size x entry_bytes)   server work

The DPF key size is the real wire size of this framework's keys
(2096 bytes, constant in n — gpudpf key format), unlike the reference's
analytic 16*4*log2(n) model (batch_pir_optimization.py:85-88); both are
available via PIRConfig.key_size_model.
"""

import collections
import hashlib
import json
import os
from typing import Dict, NamedTuple, Optional, Sequence


class HotColdConfig(NamedTuple):
    hot_fraction: float = 0.0       # fraction of table kept on-device
    hot_queries: int = 0            # unused slot kept for sweep parity


class CollocateConfig(NamedTuple):
    group_size: int = 1             # entries packed per row (1 = off)
    cache_path: Optional[str] = None  # JSON cache for the grouping


class PIRConfig(NamedTuple):
    num_bins: int = 16
    queries_per_bin: int = 1
    key_size_model: str = "gpudpf"  # "gpudpf" (2096 B const), "compact"
                                    # (16*(3+4*depth) B), or "logn"


KEY_BYTES = 2096


def _stable_hash(x: int, salt: int = 0) -> int:
    h = hashlib.blake2b(b"%d:%d" % (salt, x), digest_size=8).digest()
    return int.from_bytes(h, "little")


def pareto_front(points: Sequence[Sequence[float]], maximize_y=True):
    """Indices of the Pareto-efficient points for (cost, quality) pairs:
    minimal x, maximal y (the reference's is_pareto_efficient_simple
    analog)."""
    idx = sorted(range(len(points)), key=lambda i: (points[i][0], -points[i][1]))
    front, best = [], None
    for i in idx:
        y = points[i][1]
        if best is None or (y > best if maximize_y else y < best):
            front.append(i)
            best = y
    return front


class BatchPIROptimize:
    """Plan and simulate batched private access to an n-entry table.

    Parameters
    ----------
    num_entries : table size
    access_patterns : list of index lists (training traces) used to
        estimate frequencies and co-occurrence
    hotcold, collocate, pir : config tuples
    """

    def __init__(self, num_entries: int,
                 access_patterns: Sequence[Sequence[int]],
                 hotcold: HotColdConfig = HotColdConfig(),
                 collocate: CollocateConfig = CollocateConfig(),
                 pir: PIRConfig = PIRConfig()):
        self.num_entries = int(num_entries)
        self.hotcold = hotcold
        self.collocate = collocate
        self.pir = pir

        self._init_frequencies(access_patterns)
        self._init_hotcold()
        self._init_collocation(access_patterns)
        self._init_bins()

    # -- structure construction -------------------------------------------
    def _init_frequencies(self, patterns):
        freq = collections.Counter()
        for p in patterns:
            freq.update(p)
        self.freq = freq
        self.by_freq = sorted(range(self.num_entries),
                              key=lambda i: (-freq.get(i, 0), i))

    def _init_hotcold(self):
        n_hot = int(self.num_entries * self.hotcold.hot_fraction)
        self.hot_set = frozenset(self.by_freq[:n_hot])
        self.cold_entries = [i for i in range(self.num_entries)
                             if i not in self.hot_set]

    def _init_collocation(self, patterns):
        """Greedy frequency-ordered grouping driven by co-occurrence: each
        cold entry joins the group (of size < group_size) with which it
        co-occurs most.  Results are cached as JSON keyed by config."""
        g = self.collocate.group_size
        if g <= 1:
            self.group_of = {e: e for e in self.cold_entries}
            self.groups = {e: [e] for e in self.cold_entries}
            return
        cache_key = None
        if self.collocate.cache_path:
            cache_key = "n%d_g%d_h%.4f" % (self.num_entries, g,
                                           self.hotcold.hot_fraction)
            if os.path.exists(self.collocate.cache_path):
                with open(self.collocate.cache_path) as f:
                    cache = json.load(f)
                if cache_key in cache:
                    self.group_of = {int(k): v for k, v in
                                     cache[cache_key].items()}
                    self._groups_from_map()
                    return
        # co-occurrence counts between cold entries (sampled)
        co = collections.Counter()
        for p in patterns[:20000]:
            cold = [e for e in set(p) if e not in self.hot_set]
            cold = cold[:64]
            for i, a in enumerate(cold):
                for b in cold[i + 1:]:
                    co[(a, b) if a < b else (b, a)] += 1
        # greedy: walk entries by frequency; attach to best open group
        group_of: Dict[int, int] = {}
        group_fill = collections.Counter()
        neighbors = collections.defaultdict(list)
        for (a, b), c in co.items():
            neighbors[a].append((c, b))
            neighbors[b].append((c, a))
        for e in self.by_freq:
            if e in self.hot_set or e in group_of:
                continue
            group_of[e] = e
            group_fill[e] = 1
            for _, nb in sorted(neighbors[e], reverse=True):
                if group_fill[e] >= g:
                    break
                if nb not in group_of and nb not in self.hot_set:
                    group_of[nb] = e
                    group_fill[e] += 1
        for e in self.cold_entries:
            group_of.setdefault(e, e)
        self.group_of = group_of
        self._groups_from_map()
        if self.collocate.cache_path and cache_key:
            cache = {}
            if os.path.exists(self.collocate.cache_path):
                with open(self.collocate.cache_path) as f:
                    cache = json.load(f)
            cache[cache_key] = {str(k): v for k, v in group_of.items()}
            with open(self.collocate.cache_path, "w") as f:
                json.dump(cache, f)

    def _groups_from_map(self):
        groups = collections.defaultdict(list)
        for e, gid in self.group_of.items():
            groups[gid].append(e)
        self.groups = dict(groups)

    def _init_bins(self):
        """Hash group-ids into bins; each bin is an independent PIR table."""
        self.bin_of = {gid: _stable_hash(gid) % self.pir.num_bins
                       for gid in self.groups}
        bins = collections.defaultdict(list)
        for gid, b in self.bin_of.items():
            bins[b].append(gid)
        self.bins = dict(bins)
        self.bin_sizes = {b: len(v) for b, v in self.bins.items()}

    # -- simulation --------------------------------------------------------
    def fetch(self, indices: Sequence[int]):
        """Simulate one batch: which of `indices` are available after hot
        hits + at most queries_per_bin PIR fetches per bin (greedy:
        most-requested groups first)?  Returns (recovered_set, stats)."""
        recovered = set()
        needed_groups = collections.Counter()
        want = set(indices)
        for i in want:
            if i in self.hot_set:
                recovered.add(i)
            else:
                needed_groups[self.group_of[i]] += 1
        per_bin = collections.defaultdict(list)
        for gid, cnt in needed_groups.items():
            per_bin[self.bin_of[gid]].append((cnt, gid))
        queries = 0
        for b, cands in per_bin.items():
            cands.sort(reverse=True)
            for cnt, gid in cands[: self.pir.queries_per_bin]:
                queries += 1
                for e in self.groups[gid]:
                    if e in want:
                        recovered.add(e)
        # budgeted queries are issued even when idle in a bin (privacy:
        # the servers must not learn which bins were needed)
        total_queries = self.pir.num_bins * self.pir.queries_per_bin
        stats = {
            "requested": len(want),
            "recovered": len(recovered),
            "queries_used": queries,
            "queries_total": total_queries,
        }
        return recovered, stats

    # -- cost model --------------------------------------------------------
    def key_bytes(self, bin_entries: int) -> int:
        import math

        if self.pir.key_size_model == "gpudpf":
            return KEY_BYTES
        depth = max(7, math.ceil(math.log2(max(128, bin_entries))))
        if self.pir.key_size_model == "compact":
            # gpudpf compact wire form (DPF.key_compact): 16*(3+4*depth) B
            return 16 * (3 + 4 * depth)
        return 16 * 4 * max(1, math.ceil(math.log2(max(2, bin_entries))))

    def communication_bytes(self, entry_bytes: int = 64) -> int:
        """Per-batch client<->servers bytes: every bin gets its full query
        budget (2 keys per query, 2 responses of one row)."""
        total = 0
        row_bytes = entry_bytes * max(1, self.collocate.group_size)
        for b in range(self.pir.num_bins):
            sz = self.bin_sizes.get(b, 1)
            total += self.pir.queries_per_bin * (
                2 * self.key_bytes(sz) + 2 * row_bytes)
        return total

    def computation_entries(self) -> int:
        """Per-batch server work in table-entries scanned (each query
        expands over its whole bin)."""
        return sum(self.bin_sizes.get(b, 0) * self.pir.queries_per_bin
                   for b in range(self.pir.num_bins))

    def hot_storage_entries(self) -> int:
        return len(self.hot_set)

    # -- evaluation --------------------------------------------------------
    def evaluate(self, patterns: Sequence[Sequence[int]]):
        """Recovery statistics over evaluation traces."""
        tot_req = tot_rec = 0
        for p in patterns:
            _, s = self.fetch(p)
            tot_req += s["requested"]
            tot_rec += s["recovered"]
        return {
            "recovery_rate": tot_rec / max(1, tot_req),
            "requested": tot_req,
            "recovered": tot_rec,
        }

    def evaluate_real(self, dataset):
        """Run the workload's own accuracy metric under this PIR plan.
        `dataset` must expose evaluate(optimizer) -> dict (see
        pir/datasets/*)."""
        return dataset.evaluate(self)

    def summarize(self, entry_bytes: int = 64):
        return {
            "num_entries": self.num_entries,
            "hot_fraction": self.hotcold.hot_fraction,
            "group_size": self.collocate.group_size,
            "num_bins": self.pir.num_bins,
            "queries_per_bin": self.pir.queries_per_bin,
            "communication_bytes": self.communication_bytes(entry_bytes),
            "computation_entries": self.computation_entries(),
            "hot_storage_entries": self.hot_storage_entries(),
        }
