"""Batch-PIR co-design tests (CPU only)."""

import random

import pytest

from pir import BatchPIROptimize, CollocateConfig, HotColdConfig, PIRConfig


def _patterns(n, count, size, seed=0):
    rng = random.Random(seed)
    # zipf-ish: low indices more frequent
    return [[min(n - 1, int(rng.random() ** 2 * n)) for _ in range(size)]
            for _ in range(count)]


def test_hot_entries_always_recovered():
    n = 1024
    pats = _patterns(n, 200, 16)
    opt = BatchPIROptimize(n, pats, hotcold=HotColdConfig(hot_fraction=1.0))
    rec, stats = opt.fetch(pats[0])
    assert stats["recovered"] == stats["requested"]


def test_binning_budget_limits_recovery():
    n = 4096
    pats = _patterns(n, 100, 64, seed=3)
    opt = BatchPIROptimize(n, pats,
                           pir=PIRConfig(num_bins=4, queries_per_bin=1))
    rec, stats = opt.fetch(list(range(100, 164)))
    # at most num_bins x queries_per_bin x group_size recoveries
    assert stats["recovered"] <= 4 * 1 * 1
    # more bins -> strictly more or equal recovery
    opt2 = BatchPIROptimize(n, pats,
                            pir=PIRConfig(num_bins=64, queries_per_bin=1))
    _, s2 = opt2.fetch(list(range(100, 164)))
    assert s2["recovered"] >= stats["recovered"]


def test_collocation_recovers_groups():
    n = 512
    # entries 2i and 2i+1 always co-occur
    pats = [[2 * i, 2 * i + 1] for i in range(n // 2) for _ in range(3)]
    opt = BatchPIROptimize(n, pats,
                           collocate=CollocateConfig(group_size=2),
                           pir=PIRConfig(num_bins=8, queries_per_bin=1))
    rec, stats = opt.fetch([10, 11])
    # both members of a co-occurring pair should come from one fetch
    if opt.group_of[10] == opt.group_of[11]:
        assert {10, 11} <= rec


def test_cost_accounting_monotone():
    n = 2048
    pats = _patterns(n, 100, 16)
    base = BatchPIROptimize(n, pats, pir=PIRConfig(num_bins=8))
    more = BatchPIROptimize(n, pats, pir=PIRConfig(num_bins=32))
    assert more.communication_bytes() > base.communication_bytes()
    assert base.summarize()["computation_entries"] > 0


def test_evaluate_recovery_rate_bounds():
    n = 1024
    pats = _patterns(n, 50, 32, seed=9)
    opt = BatchPIROptimize(n, pats,
                           hotcold=HotColdConfig(hot_fraction=0.1),
                           pir=PIRConfig(num_bins=16, queries_per_bin=2))
    r = opt.evaluate(pats[:10])
    assert 0.0 < r["recovery_rate"] <= 1.0


def test_lm_dataset_degradation():
    from pir.datasets import language_model

    ds = language_model.initialize(vocab=256, corpus_len=6000, batch_size=4)
    ds.train_model(epochs=1, max_batches=5)
    clean = ds.evaluate(None, max_batches=5)
    opt = BatchPIROptimize(ds.num_entries, ds.train_patterns[:500],
                           pir=PIRConfig(num_bins=4, queries_per_bin=1))
    degraded = ds.evaluate(opt, max_batches=5)
    assert "recovery_rate" in degraded
    assert degraded["value"] >= 0  # ppl defined
    # full-budget plan should be no worse than the tiny-budget plan
    opt_full = BatchPIROptimize(ds.num_entries, ds.train_patterns[:500],
                                hotcold=HotColdConfig(hot_fraction=1.0))
    full = ds.evaluate(opt_full, max_batches=5)
    assert abs(full["value"] - clean["value"]) < 1e-6


def test_movielens_dataset():
    from pir.datasets import movielens

    ds = movielens.initialize(num_items=256, num_users=200)
    ds.train_model(epochs=1)
    res = ds.evaluate(None)
    assert res["metric"] == "auc"
    opt = BatchPIROptimize(ds.num_entries, ds.train_patterns,
                           pir=PIRConfig(num_bins=8))
    res2 = ds.evaluate(opt)
    assert 0 <= res2["recovery_rate"] <= 1


def test_taobao_dataset():
    from pir.datasets import taobao

    ds = taobao.initialize(num_items=256, num_samples=300)
    ds.train_model(epochs=1)
    opt = BatchPIROptimize(ds.num_entries, ds.train_patterns,
                           pir=PIRConfig(num_bins=8))
    res = ds.evaluate(opt)
    assert res["metric"] == "auc"
    assert 0 <= res["recovery_rate"] <= 1


def test_pareto_front():
    from pir.batch_pir import pareto_front

    pts = [(1, 1), (2, 3), (3, 2), (4, 4), (2, 0.5)]
    front = pareto_front(pts)
    assert 0 in front and 1 in front and 3 in front
    assert 2 not in front and 4 not in front
