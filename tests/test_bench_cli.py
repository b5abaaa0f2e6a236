"""bench.py contract tests: the driver launches bench.py via
torch.distributed.run, so the CLI itself (not just the library) must be
correct — these run it for real on CPU (gloo backend), including the
row-sharded strong-scaling mode with a reconstruction --check."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _last_json_line(stdout):
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError("no JSON line in bench output:\n" + stdout)


def test_bench_single_process_cpu():
    cmd = [sys.executable, "bench.py", "--device", "cpu",
           "--entries", "8192", "--batch", "8", "--steps", "2",
           "--warmup", "1", "--prf", "SALSA20", "--check"]
    r = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    rec = _last_json_line(r.stdout)
    assert rec["metric"] == "DPFs/sec"
    assert rec["value"] > 0
    assert rec["n_gpus"] == 1


@pytest.mark.parametrize("mode", ["shard", "replicate"])
def test_bench_world2_gloo_check(mode):
    """The exact launch pattern the driver uses, world=2 on CPU, with a
    reconstruction check: shard mode must produce the same values as a
    single server (the --check compares against the plain table rows)."""
    port = _free_port()
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port), "bench.py",
           "--device", "cpu", "--backend", "gloo", "--mode", mode,
           "--entries", "8192", "--batch", "8", "--steps", "2",
           "--warmup", "1", "--prf", "SALSA20", "--check"]
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    r = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                       timeout=600, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    rec = _last_json_line(r.stdout)
    assert rec["n_gpus"] == 2
    if mode == "shard":
        assert rec["scaling"] == "strong"
        assert rec["config"]["parallelism"] == "shard2-gloo"
        assert rec["config"]["global_batch"] == 8
    else:
        assert rec["scaling"] == "weak"
        assert rec["config"]["global_batch"] == 16


def test_bench_single_process_cpu_wide_entries():
    # wide entry sizes must not crash the serving-mode selection
    cmd = [sys.executable, "bench.py", "--device", "cpu",
           "--entries", "4096", "--batch", "4", "--steps", "1",
           "--warmup", "0", "--entry-size", "24", "--prf", "SALSA20",
           "--check"]
    r = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    rec = _last_json_line(r.stdout)
    assert rec["config"]["entry_size"] == 24
