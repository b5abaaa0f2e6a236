"""The dict-line -> CSV scraping contract (the reference's scrape.py
analog): benchmark harnesses print one python dict per run; scrape.py
collects the last dict-line of each file into a CSV."""

import csv
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_scrape_dict_lines(tmp_path):
    (tmp_path / "a.txt").write_text(
        "noise\n{'strategy': 'fused', 'n': 8192, 'dpfs_per_sec': 100.5}\n")
    (tmp_path / "b.txt").write_text(
        "{'strategy': 'bfs', 'n': 1024, 'dpfs_per_sec': 7.0}\n"
        "{'strategy': 'bfs', 'n': 2048, 'dpfs_per_sec': 9.0}\n")  # last wins
    (tmp_path / "junk.txt").write_text("no dict here\n")
    r = subprocess.run([sys.executable, "benchmarks/scrape.py",
                        str(tmp_path)], cwd=REPO, capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    rows = list(csv.DictReader(r.stdout.splitlines()))
    got = {row["strategy"]: row for row in rows}
    assert got["fused"]["dpfs_per_sec"] == "100.5"
    assert got["bfs"]["n"] == "2048"
    assert len(rows) == 2  # junk file skipped


def test_scrape_rejects_code_injection(tmp_path):
    # ast.literal_eval, not eval: expressions must not execute
    (tmp_path / "evil.txt").write_text(
        "{'x': __import__('os').system('touch %s')}\n"
        % (tmp_path / "pwned"))
    r = subprocess.run([sys.executable, "benchmarks/scrape.py",
                        str(tmp_path)], cwd=REPO, capture_output=True,
                       text=True, timeout=120)
    # the expression is rejected (not evaluated), leaving zero rows
    assert r.returncode == 1 and "no results" in r.stderr
    assert not os.path.exists(tmp_path / "pwned")
