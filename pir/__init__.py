"""Batch-PIR / ML co-design toolkit (capability parity with the
reference's paper/experimental tree, re-implemented from scratch).

Components:
  batch_pir   - BatchPIROptimize: hot/cold split, entry collocation,
                hash binning, greedy batch-PIR recovery simulation, and
                communication/computation cost accounting
  datasets    - PIR-degradable eval workloads (language model, MovieLens-
                style recommender, Taobao-style CTR) with synthetic-data
                fallbacks (this environment has no dataset downloads)
  sweep       - config-grid sweep driver writing one JSON per config
  codesign    - join accuracy sweeps with GPU kernel perf sweeps into
                latency/throughput/accuracy operating points
  plots       - Pareto-front extraction and plotting
"""

from pir.batch_pir import (  # noqa: F401
    BatchPIROptimize,
    CollocateConfig,
    HotColdConfig,
    PIRConfig,
)
