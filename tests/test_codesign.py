"""Codesign join tests: constraint filtering, perf-row matching, Pareto
marking (pure CPU)."""

from pir.codesign import join


def _acc(comm, num_bins=8, qpb=1, ppl=50.0, n=16384):
    return {
        "hot_fraction": 0.0, "group_size": 1,
        "num_bins": num_bins, "queries_per_bin": qpb,
        "num_entries": n, "communication_bytes": comm,
        "accuracy": {"metric": "ppl", "value": ppl},
    }


def _perf(n, thr, lat):
    return {"num_entries": float(n), "batch_size": 512.0, "prf": "AES128",
            "strategy": "fused", "throughput_dpfs_per_sec": thr,
            "latency_ms": lat}


def test_join_filters_comm_and_latency():
    acc = [_acc(comm=100_000), _acc(comm=400_000)]  # 2nd exceeds budget
    perf = [_perf(16384, 1e6, 0.5)]
    pts = join(acc, perf, max_latency_ms=100, max_comm_bytes=300_000)
    assert len(pts) == 1
    # latency filter
    perf_slow = [_perf(16384, 1e6, 500.0)]
    assert join(acc, perf_slow, max_latency_ms=100) == []


def test_join_picks_smallest_sufficient_table():
    acc = [_acc(comm=1000, num_bins=8, n=16384)]  # bin ~2048 -> needs >=2048
    perf = [_perf(1 << 20, 1e4, 1.0), _perf(4096, 1e6, 0.2),
            _perf(128, 1e7, 0.1)]
    pts = join(acc, perf)
    assert pts[0]["perf_row"]["num_entries"] == 4096


def test_join_pareto_marks_lower_ppl_and_higher_throughput():
    acc = [_acc(comm=1000, num_bins=8, ppl=40.0),
           _acc(comm=1000, num_bins=16, ppl=30.0),
           _acc(comm=1000, num_bins=16, qpb=2, ppl=35.0)]
    perf = [_perf(16384, 1e6, 0.5)]
    pts = join(acc, perf)
    assert len(pts) == 3
    # the 16-bin/qpb1 config dominates the 16-bin/qpb2 one (fewer queries ->
    # more batches/sec AND lower ppl)
    dominated = [p for p in pts
                 if p["config"]["num_bins"] == 16
                 and p["config"]["queries_per_bin"] == 2]
    assert dominated and not dominated[0]["pareto"]
    best = [p for p in pts
            if p["config"]["num_bins"] == 16
            and p["config"]["queries_per_bin"] == 1]
    assert best and best[0]["pareto"]


def test_plot_accuracy_and_codesign_smoke(tmp_path):
    """Plotters render from committed sweep/join artifacts."""
    import os

    from pir import plots

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sweep_dir = os.path.join(repo, "profiles", "codesign", "lm_sweep")
    join_json = os.path.join(repo, "profiles", "codesign",
                             "codesign_lm_r2.json")
    out1 = str(tmp_path / "acc.png")
    out2 = str(tmp_path / "cd.png")
    plots.plot_accuracy(sweep_dir, out1)
    plots.plot_codesign(join_json, out2)
    assert os.path.getsize(out1) > 1000
    assert os.path.getsize(out2) > 1000
